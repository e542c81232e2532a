"""sat_amd.ops — the op layer.

Every compute op the decoder/loss/optimizer hot path needs exists twice:

  * `sat_amd.ops.functional` — plain-PyTorch fp32 reference implementations.
    Used on CPU, and as the ground truth the HIP kernels are tested against
    (tests/test_ops_gpu.py).
  * `sat_amd.ops.hip` — hand-written CDNA4 HIP kernels (MFMA GEMM tiles,
    LDS-staged attention softmax, fused LSTM gates, fused CE, fused Adam)
    exposed through torch.autograd.Functions.

Dispatch rule (no multi-backend sprawl): CUDA/HIP tensors take the HIP kernel
path; on a GPU box a missing extension raises instead of silently falling
back to eager PyTorch.  CPU tensors take the reference path.
"""

import torch

from . import functional as F  # noqa: N812


def _use_hip(*tensors):
    """GPU bf16 tensors -> HIP kernels (hard error if the extension is
    missing); CPU -> reference path.  fp32 on GPU is the explicit
    `compute_dtype='fp32'` debug knob and runs eager PyTorch-ROCm."""
    primary = tensors[0]
    if not primary.is_cuda:
        return False
    if primary.dtype != torch.bfloat16:
        return False
    from . import hip
    hip.require()  # raises if the extension is missing on a GPU box
    return True


def dense(x, weight, bias=None, activation=None):
    """act(x @ weight.T + bias). x: [M,K], weight: [N,K] (torch layout)."""
    if _use_hip(x, weight) and x.shape[-1] % 8 == 0:
        from . import hip
        return hip.dense(x, weight, bias, activation)
    return F.dense(x, weight, bias, activation)


def lstm_cell(x, h, c, weight, bias, forget_bias=1.0):
    """One TF-semantics LSTMCell step; returns (h', c')."""
    if _use_hip(x, weight) and (x.shape[1] + h.shape[1]) % 8 == 0:
        from . import hip
        return hip.lstm_cell(x, h, c, weight, bias, forget_bias)
    return F.lstm_cell(x, h, c, weight, bias, forget_bias)


def attention_pool(contexts, logits):
    """softmax over L + weighted context sum; returns (alpha, context)."""
    if _use_hip(contexts) and contexts.shape[1] <= 1024 \
            and contexts.shape[2] % 8 == 0:
        from . import hip
        return hip.attention_pool(contexts, logits)
    return F.attention_pool(contexts, logits)


def attention_tail(t1, t2, v, contexts, p, training, seed_dev, salt):
    """Fused attention tail: dropout(t1 + tiled t2) · v -> softmax over L
    -> weighted context sum.  Replaces the reference's tile/add/dropout/
    N=1-GEMM/softmax/weighted-sum chain (model.py:425-435, :263-264) with
    two fused kernels on GPU; counter-based dropout (seed_dev, salt) keeps
    it hipGraph-safe."""
    if (_use_hip(contexts, t1) and t1.shape[1] % 512 == 0
            and t1.shape[1] <= 2048 and contexts.shape[1] <= 1024
            and contexts.shape[2] % 8 == 0):
        from . import hip
        return hip.attention_tail(t1, t2, v, contexts,
                                  p if training else 0.0, seed_dev, salt)
    return F.attention_tail(t1, t2, v, contexts, p, training)


def embedding(ids, table):
    if _use_hip(table):
        from . import hip
        return hip.embedding(ids, table)
    return F.embedding(ids, table)


def masked_softmax_ce(logits, labels, mask):
    """Per-element masked CE ([B]) — sum/Σmask is done by the caller."""
    if _use_hip(logits):
        from . import hip
        return hip.masked_softmax_ce(logits, labels, mask)
    return F.masked_softmax_ce(logits, labels, mask)


def dropout(x, rate, training):
    return F.dropout(x, rate, training)
