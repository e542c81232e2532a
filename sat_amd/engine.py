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
        self.mode = None       # 'full' | 'split' once captured
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
        if hasattr(opt, 'sync_shadows'):
            opt.sync_shadows()  # restored params -> refresh bf16 shadows
        rng = self._rng_tensor()
        if rng is not None and 'rng' in snap:
            rng.copy_(snap['rng'])

    def _inner_nocomm(self, images, sentences, masks):
        """Forward + backward only — the capturable portion of the
        split mode (collectives and optimizer stay outside)."""
        out = self.model(images, sentences, masks)
        opt = self.optimizer
        fused = (opt.kind == 'Adam' and len(opt.params) > 0
                 and opt.params[0].is_cuda)
        if not fused:
            for p in opt.params:
                if p.grad is not None:
                    p.grad.zero_()
        out['total_loss'].backward()
        return out

    def _split_tail(self):
        self.ddp.launch_deferred_comm()
        self.ddp.wait_deferred_comm()
        self.optimizer.step()

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

    def _capture_split(self, images, sentences, masks):
        """Split mode: capture fwd+bwd (including the DDP bucket
        fill/scale kernels) in one graph; RCCL collectives + the fused
        Adam run eagerly after each replay.  Used when full capture
        fails at world > 1 (e.g. a collective that cannot be captured):
        keeps ~99% of the launches in the graph without putting RCCL
        inside it."""
        self.ddp.defer_comm = True
        self.static_in = (images.clone(), sentences.clone(), masks.clone())
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(self.warmup_iters):
                self._inner_nocomm(*self.static_in)
                self.ddp.snapshot_capture_state()
                self._split_tail()
                self.ddp.reset()
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_out = self._inner_nocomm(*self.static_in)
        torch.cuda.synchronize()
        self.ddp.snapshot_capture_state()

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

    def _try_mode(self, mode, images, sentences, masks):
        """Attempt one capture mode with cross-rank agreement; returns
        True if ALL ranks succeeded (graph ready)."""
        snap = self._snapshot()
        ok = True
        try:
            if mode == 'full':
                self._capture(images, sentences, masks)
            else:
                self._capture_split(images, sentences, masks)
        except Exception as e:
            print('[sat_amd] hipGraph %s-capture failed (%r)'
                  % (mode, e))
            ok = False
        agreed = self._all_ranks_ok(ok)
        if ok and not agreed:
            print('[sat_amd] hipGraph %s-capture failed on a peer rank'
                  % mode)
        torch.cuda.synchronize()
        self._restore(snap)
        if not agreed:
            self.graph = None
            if self.ddp is not None:
                self.ddp.defer_comm = False
                self.ddp.reset()
        return agreed

    def step(self, images, sentences, masks):
        if self.failed:
            return self._inner(images, sentences, masks)
        if self.graph is None and self.mode is None:
            # warmup/capture run real optimizer steps on the first
            # batch; _try_mode snapshots/restores params, Adam state,
            # step count and RNG around every attempt, and all ranks
            # agree on the outcome before anyone replays (mismatched
            # per-step collective counts would hang the job).
            if self._try_mode('full', images, sentences, masks):
                self.mode = 'full'
            elif self.ddp is not None and self._try_mode(
                    'split', images, sentences, masks):
                self.mode = 'split'
            else:
                print('[sat_amd] all capture modes failed; '
                      'running eager steps')
                self.failed = True
                return self._inner(images, sentences, masks)
        si, ss, sm = self.static_in
        si.copy_(images, non_blocking=True)
        ss.copy_(sentences, non_blocking=True)
        sm.copy_(masks, non_blocking=True)
        self.graph.replay()
        if self.mode == 'split':
            self._split_tail()
        else:
            self.optimizer.step_count += 1
        return self.static_out
