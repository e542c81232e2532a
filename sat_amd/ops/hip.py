"""HIP kernel path: autograd wrappers over the sat_amd._C extension.

The extension (built by setup.py / __graft_entry__.build() from
sat_amd/ops/csrc/*.hip, gfx950-only) provides:

  dense_fwd(x, w, bias, act)            MFMA GEMM + fused bias/activation
  lstm_pointwise_fwd / _bwd             fused LSTM gate math (i,j,f,o)
  attn_softmax_ctx_fwd / _bwd           LDS-staged softmax over L + ctx-sum
  embedding_fwd / embedding_bwd         gather / scatter-add
  ce_fwd / ce_bwd                       fused masked softmax cross-entropy
  grad_sq_norm / adam_step              fused global-norm clip + Adam

Backward GEMMs (dX = dY Wᵀ, dW = Xᵀ dY) go through torch.matmul, i.e.
hipBLASLt — plain library GEMMs, per the MI355X design rules; everything
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


class _Dense(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, act):
        y = _C.dense_fwd(x, w, b if b is not None else torch.Tensor(),
                         _ACT[act])
        ctx.save_for_backward(x, w, y)
        ctx.act = act
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        if ctx.act == 'tanh':
            dpre = dy * (1 - y.float() * y.float()).to(dy.dtype)
        elif ctx.act == 'relu':
            dpre = dy * (y > 0).to(dy.dtype)
        else:
            dpre = dy
        dx = dpre.matmul(w.t()) if ctx.needs_input_grad[0] else None
        dw = x.t().matmul(dpre) if ctx.needs_input_grad[1] else None
        db = dpre.sum(0) if (ctx.has_bias and ctx.needs_input_grad[2]) \
            else None
        return dx, dw, db, None


def dense(x, weight, bias=None, activation=None):
    return _Dense.apply(x, weight, bias, activation)


class _LSTMPointwise(torch.autograd.Function):
    """gates [B,4H] + c [B,H] -> (h', c'); saves sigmoid/tanh activations."""

    @staticmethod
    def forward(ctx, gates, c, forget_bias):
        h_new, c_new, saved = _C.lstm_pointwise_fwd(gates, c, forget_bias)
        ctx.save_for_backward(c, c_new, saved)
        return h_new, c_new

    @staticmethod
    def backward(ctx, dh, dc):
        c, c_new, saved = ctx.saved_tensors
        dgates, dc_prev = _C.lstm_pointwise_bwd(
            dh.contiguous(), dc.contiguous(), c, c_new, saved)
        return dgates, dc_prev, None


def lstm_cell(x, h, c, weight, bias, forget_bias=1.0):
    xh = torch.cat([x, h], dim=1)
    gates = dense(xh, weight, bias, None)
    return _LSTMPointwise.apply(gates, c, forget_bias)


class _AttnSoftmaxCtx(torch.autograd.Function):
    @staticmethod
    def forward(ctx, contexts, logits):
        alpha, pooled = _C.attn_softmax_ctx_fwd(contexts, logits)
        ctx.save_for_backward(contexts, alpha)
        return alpha, pooled

    @staticmethod
    def backward(ctx, dalpha, dpooled):
        contexts, alpha = ctx.saved_tensors
        dctx, dlogits = _C.attn_softmax_ctx_bwd(
            contexts, alpha, dalpha.contiguous(), dpooled.contiguous())
        return dctx, dlogits


def attention_pool(contexts, logits):
    return _AttnSoftmaxCtx.apply(contexts, logits)


class _Embedding(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, table):
        ctx.save_for_backward(ids)
        ctx.rows = table.shape[0]
        ctx.table_dtype = table.dtype
        return _C.embedding_fwd(ids, table)

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
        losses, lse = _C.ce_fwd(logits, labels, mask)
        ctx.save_for_backward(logits, labels, mask, lse)
        return losses

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, mask, lse = ctx.saved_tensors
        dlogits = _C.ce_bwd(logits, labels, mask, lse, dloss.contiguous())
        return dlogits, None, None


def masked_softmax_ce(logits, labels, mask):
    return _MaskedCE.apply(logits, labels, mask)


# ---- fused optimizer (no autograd; called by sat_amd.optim) ----

def grad_sq_norm(grads):
    return _C.grad_sq_norm(list(grads))


def adam_step(params, grads, ms, vs, lr, beta1, beta2, eps, step,
              clip, grad_sq):
    """Fused multi-tensor Adam. `grad_sq` is the on-device Σ‖g‖² scalar from
    grad_sq_norm; the kernel derives scale = clip/max(clip, √grad_sq)."""
    _C.adam_step(list(params), list(grads), list(ms), list(vs),
                 float(lr), float(beta1), float(beta2), float(eps),
                 int(step), float(clip), grad_sq)
