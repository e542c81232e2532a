"""hipGraph-captured training step.

The reference amortizes per-op dispatch by building one static TF graph and
letting the C++ runtime execute it.  The MI355X-native
equivalent is hipGraph capture: the whole training step — CNN forward,
20-step attention-LSTM forward, backward, bucketed all-reduce (DP) and the
fused Adam — is captured once into a hipGraph and replayed per step, so the
~1500 kernel launches cost one graph launch instead of ~1500 dispatches.

Requirements made true elsewhere:
  * the fused optimizer keeps its step counter / bias correction / LR decay
    on device (sat_amd/optim.py), so replays see advancing state;
  * gradients are zeroed in-place inside the captured region (never
    `set_to_none`);
  * input batches are copied into static device buffers before replay.

If capture fails (e.g. an op not capture-safe under a given world size),
the engine falls back to eager execution with a warning — numerics are
identical either way.
"""

import torch


class GraphedTrainStep(object):
    def __init__(self, model, optimizer, ddp=None, warmup_iters=3):
        self.model = model
        self.optimizer = optimizer
        self.ddp = ddp
        self.warmup_iters = warmup_iters
        self.graph = None
        self.failed = False
        self.static_in = None
        self.static_out = None

    def _inner(self, images, sentences, masks):
        out = self.model(images, sentences, masks)
        opt = self.optimizer
        fused = (opt.kind == 'Adam' and len(opt.params) > 0
                 and opt.params[0].is_cuda)
        if not fused:
            # the fused Adam zeroes grads inside its own update kernel
            for p in opt.params:
                if p.grad is not None:
                    p.grad.zero_()
        out['total_loss'].backward()
        if self.ddp is not None:
            self.ddp.finish_backward()
        self.optimizer.step()
        return out

    def _rng_tensor(self):
        dec = getattr(self.model, 'decoder', None)
        return getattr(dec, '_rng', None) if dec is not None else None

    def _snapshot(self):
        opt = self.optimizer
        snap = {
            'params': [p.detach().clone() for p in opt.params],
            'state': {i: {k: t.clone() for k, t in opt.state[p].items()}
                      for i, p in enumerate(opt.params)},
            'step_count': opt.step_count,
        }
        rng = self._rng_tensor()
        if rng is not None:
            snap['rng'] = rng.clone()
        return snap

    def _restore(self, snap):
        opt = self.optimizer
        with torch.no_grad():
            for p, sp in zip(opt.params, snap['params']):
                p.copy_(sp)
            for i, p in enumerate(opt.params):
                for k, t in opt.state[p].items():
                    t.copy_(snap['state'][i][k])
        opt.step_count = snap['step_count']
        if getattr(opt, 'step_dev', None) is not None:
            opt.step_dev.fill_(float(opt.step_count))
        rng = self._rng_tensor()
        if rng is not None and 'rng' in snap:
            rng.copy_(snap['rng'])

    def _capture(self, images, sentences, masks):
        self.static_in = (images.clone(), sentences.clone(), masks.clone())
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(self.warmup_iters):
                self._inner(*self.static_in)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_out = self._inner(*self.static_in)
        torch.cuda.synchronize()

    def _all_ranks_ok(self, ok):
        """Agree across ranks on the capture outcome so no rank replays a
        graph while another runs eager (their per-step collective counts
        would then diverge and the job would hang)."""
        import torch.distributed as dist
        if not (dist.is_available() and dist.is_initialized()
                and dist.get_world_size() > 1):
            return ok
        dev = 'cpu' if dist.get_backend() == 'gloo' else 'cuda'
        flag = torch.tensor([1.0 if ok else 0.0], device=dev)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN)
        return bool(flag.item() >= 0.5)

    def step(self, images, sentences, masks):
        if self.failed:
            return self._inner(images, sentences, masks)
        if self.graph is None:
            # warmup/capture run real optimizer steps on the first batch;
            # snapshot HERE (not inside _capture) so a partial capture
            # failure still restores params/Adam state/step count/RNG
            # before the eager fallback — otherwise the warmup steps'
            # duplicate updates would be silently kept.
            snap = self._snapshot()
            ok = True
            try:
                self._capture(images, sentences, masks)
            except Exception as e:
                print('[sat_amd] hipGraph capture failed (%r); '
                      'falling back to eager steps' % (e,))
                ok = False
            if not self._all_ranks_ok(ok):
                if ok:
                    print('[sat_amd] hipGraph capture failed on a peer '
                          'rank; all ranks falling back to eager steps')
                self.failed = True
                self.graph = None
            torch.cuda.synchronize()
            self._restore(snap)
            if self.failed:
                return self._inner(images, sentences, masks)
        si, ss, sm = self.static_in
        si.copy_(images, non_blocking=True)
        ss.copy_(sentences, non_blocking=True)
        sm.copy_(masks, non_blocking=True)
        self.graph.replay()
        self.optimizer.step_count += 1
        return self.static_out
