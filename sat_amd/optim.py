"""Optimizers + global-norm gradient clipping (parity with reference
`model.py:461-513` build_optimizer).

Supports 'Adam' (default; β1/β2/ε from config, config.py:41-43), 'RMSProp',
'Momentum' (with use_nesterov) and 'SGD', all behind global-norm clipping at
config.clip_gradients (model.py:510) and the optional staircase exponential
LR decay (model.py:466-474, inactive at the default decay factor 1.0).

On GPU the Adam path runs two hand-written CDNA4 kernels per step
(sat_amd/ops/csrc/adam.hip): a multi-tensor grad-norm² reduction and a
multi-tensor clip+Adam update — one launch each over all ~14M decoder params
instead of per-tensor eager ops.  Other optimizers use PyTorch-ROCm eager
math (parity features, not the flagship path).
"""

import math

import torch


class Optimizer(object):
    def __init__(self, config, params):
        self.config = config
        self.params = [p for p in params if p.requires_grad]
        self.kind = config.optimizer
        self.step_count = 0
        self.state = {}
        for p in self.params:
            s = {}
            if self.kind == 'Adam':
                s['m'] = torch.zeros_like(p, dtype=torch.float32)
                s['v'] = torch.zeros_like(p, dtype=torch.float32)
            elif self.kind == 'RMSProp':
                s['ms'] = torch.zeros_like(p, dtype=torch.float32)
                s['mom'] = torch.zeros_like(p, dtype=torch.float32)
                if config.centered:
                    s['mg'] = torch.zeros_like(p, dtype=torch.float32)
            elif self.kind == 'Momentum':
                s['mom'] = torch.zeros_like(p, dtype=torch.float32)
            self.state[p] = s

    def learning_rate(self):
        cfg = self.config
        lr = cfg.initial_learning_rate
        if cfg.learning_rate_decay_factor < 1.0:
            # staircase exponential decay (model.py:466-474)
            lr = lr * cfg.learning_rate_decay_factor ** (
                self.step_count // cfg.num_steps_per_decay)
        return lr

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    # ---- bf16 shadow weights ----
    # The fused Adam refreshes persistent bf16 copies of registered
    # params in its own update pass (free — the new value is already in
    # a register), so the compute path reads shadows instead of casting
    # weights every forward.  Non-fused/eager steps fall back to an
    # explicit sync so shadows can never go stale.

    def register_shadows(self, mapping):
        """mapping: {param: bf16 tensor of the same shape}."""
        self._shadows = dict(mapping)
        self._mt_ptrs = None  # force desc rebuild
        self.sync_shadows()

    def sync_shadows(self):
        for p, sh in getattr(self, '_shadows', {}).items():
            with torch.no_grad():
                sh.copy_(p.detach().to(sh.dtype))

    @torch.no_grad()
    def step(self):
        cfg = self.config
        self.step_count += 1
        lr = self.learning_rate()
        grads = [p.grad if p.grad is not None else torch.zeros_like(p)
                 for p in self.params]

        use_hip = (len(self.params) > 0 and self.params[0].is_cuda
                   and self.kind == 'Adam')
        if use_hip:
            from sat_amd import _C
            from .ops import hip
            hip.require()
            # Everything stays on-device (grad-norm², clip scale, step
            # counter, bias correction, LR decay): zero host syncs, so the
            # whole optimizer step is hipGraph-capturable.  All tensors go
            # through ONE multi-tensor kernel via a device pointer table,
            # rebuilt only when any data pointer changes.
            if not hasattr(self, 'step_dev') or self.step_dev is None:
                self.step_dev = torch.zeros(
                    (), dtype=torch.float32,
                    device=self.params[0].device)
                self.step_dev.fill_(float(self.step_count - 1))
            self.step_dev.add_(1.0)
            shadows = getattr(self, '_shadows', {})
            ptrs = tuple(
                (p.data.data_ptr(), g.data_ptr(),
                 self.state[p]['m'].data_ptr(),
                 self.state[p]['v'].data_ptr(),
                 shadows[p].data_ptr() if p in shadows else 0)
                for p, g in zip(self.params, grads))
            if getattr(self, '_mt_ptrs', None) != ptrs:
                dev = self.params[0].device
                self._mt_ptrs = ptrs
                self._mt_desc = torch.tensor(
                    [list(row) for row in ptrs],
                    dtype=torch.int64).to(dev)
                numels = [p.numel() for p in self.params]
                cum = [0]
                for n in numels:
                    cum.append(cum[-1] + n)
                self._mt_cum = torch.tensor(
                    cum, dtype=torch.int64).to(dev)
                self._mt_total = cum[-1]
                # float4 path needs every tensor boundary 16B-aligned
                self._mt_vec4 = all(c % 4 == 0 for c in cum)
                for g, p0 in zip(grads, self.params):
                    # dense storage in any layout (channels_last conv
                    # grads included) is fine: p/g/m/v share the layout
                    # and the kernel iterates flat storage order
                    if g.dtype != torch.float32 or not (
                            g.is_contiguous()
                            or g.is_contiguous(
                                memory_format=torch.channels_last)):
                        raise RuntimeError(
                            'fused Adam needs dense fp32 grads; got '
                            '%s for a %s param' % (g.dtype,
                                                   tuple(p0.shape)))
            gsq = _C.sq_norm_mt(self._mt_desc, self._mt_cum,
                                len(self.params), self._mt_total)
            # zero_grads=True: the update kernel clears p.grad in the
            # same pass, so the engine skips its per-tensor fill
            # launches and autograd accumulates into stable addresses
            _C.adam_step_mt(
                self._mt_desc, self._mt_cum, len(self.params),
                self._mt_total, self.step_dev,
                cfg.initial_learning_rate,
                cfg.learning_rate_decay_factor, cfg.num_steps_per_decay,
                cfg.beta1, cfg.beta2, cfg.epsilon,
                cfg.clip_gradients, gsq, True,
                getattr(self, '_mt_vec4', False))
            return

        # ---- eager path (CPU, or non-Adam optimizers) ----
        gnorm = math.sqrt(sum(float((g.float() ** 2).sum()) for g in grads))
        scale = 1.0
        if cfg.clip_gradients and gnorm > cfg.clip_gradients:
            scale = cfg.clip_gradients / (gnorm + 1e-12)

        for p, g in zip(self.params, grads):
            g = g.float() * scale
            s = self.state[p]
            if self.kind == 'Adam':
                s['m'].mul_(cfg.beta1).add_(g, alpha=1 - cfg.beta1)
                s['v'].mul_(cfg.beta2).addcmul_(g, g, value=1 - cfg.beta2)
                mhat = s['m'] / (1 - cfg.beta1 ** self.step_count)
                vhat = s['v'] / (1 - cfg.beta2 ** self.step_count)
                upd = mhat / (vhat.sqrt() + cfg.epsilon)
            elif self.kind == 'RMSProp':
                # TF RMSPropOptimizer: mom = m*mom + lr*g/sqrt(ms[-mg^2]+eps)
                # and var -= mom (epsilon inside the sqrt)
                s['ms'].mul_(cfg.decay).addcmul_(g, g, value=1 - cfg.decay)
                denom = s['ms']
                if cfg.centered:
                    s['mg'].mul_(cfg.decay).add_(g, alpha=1 - cfg.decay)
                    denom = denom - s['mg'] ** 2
                s['mom'].mul_(cfg.momentum).add_(
                    lr * g / (denom + cfg.epsilon).sqrt())
                p.data.add_(-s['mom'].to(p.dtype))
                continue
            elif self.kind == 'Momentum':
                s['mom'].mul_(cfg.momentum).add_(g)
                if cfg.use_nesterov:
                    upd = g + cfg.momentum * s['mom']
                else:
                    upd = s['mom']
            elif self.kind == 'SGD':
                upd = g
            else:
                raise ValueError('unknown optimizer %r' % (self.kind,))
            p.data.add_(upd.to(p.dtype), alpha=-lr)
        self.sync_shadows()

    # ---- checkpoint support: optimizer slots are saved like TF's ----

    def state_arrays(self):
        out = {}
        for i, p in enumerate(self.params):
            for k, t in self.state[p].items():
                out['optimizer/%d/%s' % (i, k)] = t
        out['optimizer/step_count'] = torch.tensor(
            float(self.step_count))
        return out

    def load_state_arrays(self, arrays):
        for i, p in enumerate(self.params):
            for k in self.state[p]:
                key = 'optimizer/%d/%s' % (i, k)
                if key in arrays:
                    self.state[p][k].copy_(
                        torch.as_tensor(arrays[key]).to(
                            self.state[p][k].device))
        if 'optimizer/step_count' in arrays:
            self.step_count = int(float(arrays['optimizer/step_count']))
            if getattr(self, 'step_dev', None) is not None:
                self.step_dev.fill_(float(self.step_count))
        self.sync_shadows()
