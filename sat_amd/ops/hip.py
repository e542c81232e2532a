"""HIP kernel path: autograd wrappers over the sat_amd._C extension.

The extension (built in-tree by setup.py / __graft_entry__.build() from
sat_amd/ops/csrc/*.hip, gfx950-only) provides:

  dense_fwd                       MFMA GEMM + fused bias/activation (bf16)
  lstm_pointwise_fwd / _bwd       fused LSTM gate math (i,j,f,o; recompute bwd)
  attn_pool_fwd                   LDS-staged softmax over L + ctx-weighted sum
  embedding_fwd / embedding_bwd   gather / scatter-add
  ce_fwd / ce_bwd                 fused masked softmax cross-entropy
  grad_sq_norm / adam_step        fused global-norm clip + Adam

Backward GEMMs (dX = dY·W, dW = dYᵀ·X) go through torch.matmul, i.e.
hipBLASLt — plain library GEMMs per the MI355X design rules; everything
fused is hand-written CDNA4.

On a GPU box a missing extension is a hard error (`require()`): the HIP path
must be the one that runs, never a silent eager fallback.
"""

import torch

_C = None
_IMPORT_ERR = None
try:
    from sat_amd import _C as _ext  # built in-tree by setup.py
    _C = _ext
except Exception as e:  # pragma: no cover
    _IMPORT_ERR = e


def available():
    return _C is not None


def require():
    if _C is None:
        raise RuntimeError(
            "sat_amd._C HIP extension is not built but a GPU tensor reached "
            "the op layer. Build it with `python setup.py build_ext --inplace`"
            " (PYTORCH_ROCM_ARCH=gfx950). Import error: %r" % (_IMPORT_ERR,))


_ACT = {None: 0, 'none': 0, 'tanh': 1, 'relu': 2}
_EMPTY = {}


def _empty(device):
    key = str(device)
    if key not in _EMPTY:
        _EMPTY[key] = torch.empty(0, device=device, dtype=torch.bfloat16)
    return _EMPTY[key]


class _Dense(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, act):
        y = _C.dense_fwd(x.contiguous(), w.contiguous(),
                         b.contiguous() if b is not None else _empty(x.device),
                         _ACT[act])
        ctx.save_for_backward(x, w, y)
        ctx.act = act
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        if ctx.act in ('tanh', 'relu'):
            dpre = _C.act_bwd(dy.contiguous(), y, _ACT[ctx.act])
        else:
            dpre = dy
        dx = dpre.matmul(w) if ctx.needs_input_grad[0] else None
        dw = dpre.t().matmul(x) if ctx.needs_input_grad[1] else None
        db = dpre.sum(0) if (ctx.has_bias and ctx.needs_input_grad[2]) \
            else None
        return dx, dw, db, None


def dense(x, weight, bias=None, activation=None):
    """act(x @ weightᵀ + bias); weight stored [out, in]."""
    return _Dense.apply(x, weight, bias, activation)


class _LSTMPointwise(torch.autograd.Function):
    """gates [B,4H] + c [B,H] -> (h', c'); backward recomputes activations
    from the saved gates (cheap elementwise; avoids 4 extra saved tensors)."""

    @staticmethod
    def forward(ctx, gates, c, forget_bias):
        h_new, c_new = _C.lstm_pointwise_fwd(gates, c, forget_bias)
        ctx.save_for_backward(gates, c)
        ctx.fb = forget_bias
        return h_new, c_new

    @staticmethod
    def backward(ctx, dh, dc):
        gates, c = ctx.saved_tensors
        dc_arg = dc.contiguous() if dc is not None else \
            torch.empty(0, device=gates.device, dtype=gates.dtype)
        dgates, dc_prev = _C.lstm_pointwise_bwd(
            gates, c, dh.contiguous(), dc_arg, ctx.fb)
        return dgates, dc_prev, None


def lstm_cell(x, h, c, weight, bias, forget_bias=1.0):
    xh = torch.cat([x, h], dim=1)
    gates = dense(xh, weight, bias, None)
    return _LSTMPointwise.apply(gates, c, forget_bias)


class _AttnPool(torch.autograd.Function):
    """(contexts [B,L,D] bf16, logits [B,L] fp32) -> (alpha fp32, pooled
    bf16).  Fused LDS-staged kernels both directions."""

    @staticmethod
    def forward(ctx, contexts, logits):
        alpha, pooled = _C.attn_pool_fwd(contexts, logits.contiguous())
        ctx.save_for_backward(contexts, alpha)
        return alpha, pooled

    @staticmethod
    def backward(ctx, dalpha, dpooled):
        contexts, alpha = ctx.saved_tensors
        da = dalpha.contiguous() if dalpha is not None \
            else torch.empty(0, device=alpha.device, dtype=alpha.dtype)
        dlogits, dctx = _C.attn_pool_bwd(
            contexts, alpha, da, dpooled.contiguous(),
            bool(ctx.needs_input_grad[0]))
        return (dctx if ctx.needs_input_grad[0] else None), dlogits


class _AttnTail(torch.autograd.Function):
    """The whole attention tail, fused (reference model.py:425-435 +
    :263-264): t = dropout(t1 + tiled t2), logits = t·v, α = softmax over
    L, pooled = Σ_l α_l·ctx_l.  Dropout masks are counter-based
    (seed_dev advances on device once per step -> hipGraph-safe) and
    regenerated in backward with zero mask storage."""

    @staticmethod
    def forward(ctx, t1, t2, v, contexts, p, seed_dev, salt):
        B, L = contexts.shape[0], contexts.shape[1]
        tdrop, logits = _C.attn_scores_fused(
            t1.contiguous(), t2.contiguous(), v.contiguous(), seed_dev,
            float(p), int(salt), L)
        alpha, pooled = _C.attn_pool_fwd(contexts, logits)
        ctx.save_for_backward(tdrop, v, alpha, contexts, seed_dev)
        ctx.p = float(p)
        ctx.salt = int(salt)
        ctx.L = L
        return alpha, pooled

    @staticmethod
    def backward(ctx, dalpha, dpooled):
        tdrop, v, alpha, contexts, seed_dev = ctx.saved_tensors
        da = dalpha.contiguous() if dalpha is not None \
            else torch.empty(0, device=alpha.device, dtype=alpha.dtype)
        dlogits, dctx = _C.attn_pool_bwd(
            contexts, alpha, da, dpooled.contiguous(),
            bool(ctx.needs_input_grad[3]))
        dt1, dt2f, dvf = _C.attn_scores_bwd(
            tdrop, v, dlogits, seed_dev, ctx.p, ctx.salt, ctx.L)
        return (dt1, dt2f.to(tdrop.dtype), dvf.to(v.dtype),
                (dctx if ctx.needs_input_grad[3] else None), None, None,
                None)


def attention_tail(t1, t2, v, contexts, p, seed_dev, salt):
    """t1: [B·L,A], t2: [B,A], v: [A] -> (alpha [B,L] fp32, pooled [B,D])."""
    return _AttnTail.apply(t1, t2, v, contexts, p, seed_dev, salt)


def attention_pool(contexts, logits):
    return _AttnPool.apply(contexts, logits.float())


class _Embedding(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, table):
        ids = ids.contiguous()
        ctx.save_for_backward(ids)
        ctx.rows = table.shape[0]
        ctx.table_dtype = table.dtype
        return _C.embedding_fwd(ids, table.contiguous())

    @staticmethod
    def backward(ctx, dy):
        (ids,) = ctx.saved_tensors
        dtable = _C.embedding_bwd(ids, dy.contiguous(), ctx.rows)
        return None, dtable.to(ctx.table_dtype)


def embedding(ids, table):
    return _Embedding.apply(ids, table)


class _MaskedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, mask):
        labels = labels.contiguous()
        mask = mask.contiguous()
        losses, lse = _C.ce_fwd(logits, labels, mask)
        ctx.save_for_backward(logits, labels, mask, lse)
        return losses

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, mask, lse = ctx.saved_tensors
        dlogits = _C.ce_bwd(logits, labels, mask, lse,
                            dloss.contiguous().float())
        return dlogits, None, None


def masked_softmax_ce(logits, labels, mask):
    return _MaskedCE.apply(logits.contiguous(), labels, mask)


# ---- fused optimizer (no autograd; called by sat_amd.optim) ----

def grad_sq_norm(grads):
    return _C.grad_sq_norm([g.contiguous() for g in grads])


def adam_step(params, grads, ms, vs, step_dev, lr0, decay_factor,
              steps_per_decay, beta1, beta2, eps, clip, grad_sq):
    """Fused multi-tensor Adam.  `grad_sq` is the on-device Σ‖g‖² scalar
    from grad_sq_norm (clip scale derived in-kernel); `step_dev` is the
    on-device step counter (bias correction + staircase LR decay computed
    in-kernel) so the whole optimizer step is hipGraph-capturable."""
    _C.adam_step(list(params), [g.contiguous() for g in grads],
                 list(ms), list(vs), step_dev,
                 float(lr0), float(decay_factor), float(steps_per_decay),
                 float(beta1), float(beta2), float(eps), float(clip),
                 grad_sq)
