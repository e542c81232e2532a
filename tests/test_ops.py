"""Unit tests of the op layer's reference (CPU) implementations.

These define the numerics contract the HIP kernels are validated against in
tests/test_ops_gpu.py.
"""

import numpy as np
import torch

from sat_amd.ops import functional as F


def test_dense_matches_manual():
    torch.manual_seed(0)
    x = torch.randn(4, 5)
    w = torch.randn(3, 5)
    b = torch.randn(3)
    y = F.dense(x, w, b, 'tanh')
    ref = torch.tanh(x @ w.t() + b)
    assert torch.allclose(y, ref)


def test_dense_no_activation_no_bias():
    x = torch.randn(4, 5)
    w = torch.randn(3, 5)
    assert torch.allclose(F.dense(x, w), x @ w.t())


def test_lstm_cell_matches_tf_semantics():
    """Check gate order (i,j,f,o) + forget bias against a manual formula."""
    torch.manual_seed(1)
    B, I, H = 3, 4, 5
    x = torch.randn(B, I)
    h = torch.randn(B, H)
    c = torch.randn(B, H)
    w = torch.randn(4 * H, I + H)
    b = torch.randn(4 * H)

    new_h, new_c = F.lstm_cell(x, h, c, w, b, forget_bias=1.0)

    g = torch.cat([x, h], 1) @ w.t() + b
    i, j, f, o = g[:, :H], g[:, H:2*H], g[:, 2*H:3*H], g[:, 3*H:]
    exp_c = c * torch.sigmoid(f + 1.0) + torch.sigmoid(i) * torch.tanh(j)
    exp_h = torch.tanh(exp_c) * torch.sigmoid(o)
    assert torch.allclose(new_c, exp_c, atol=1e-6)
    assert torch.allclose(new_h, exp_h, atol=1e-6)


def test_lstm_cell_matches_torch_lstmcell():
    """Cross-check against torch.nn.LSTMCell (gate order i,f,g,o there)."""
    torch.manual_seed(2)
    B, I, H = 2, 3, 4
    cell = torch.nn.LSTMCell(I, H)
    x = torch.randn(B, I)
    h = torch.randn(B, H)
    c = torch.randn(B, H)
    ht, ct = cell(x, (h, c))

    # rearrange torch's (i,f,g,o) rows into our (i,j,f,o) TF layout
    wi = cell.weight_ih  # [4H, I]
    wh = cell.weight_hh
    bias = cell.bias_ih + cell.bias_hh
    def tf_order(m):
        i, f, g, o = m.chunk(4, dim=0)
        return torch.cat([i, g, f, o], dim=0)
    w = torch.cat([tf_order(wi), tf_order(wh)], dim=1).contiguous()
    b = tf_order(bias.unsqueeze(1)).squeeze(1)
    new_h, new_c = F.lstm_cell(x, h, c, w, b, forget_bias=0.0)
    assert torch.allclose(new_h, ht, atol=1e-5)
    assert torch.allclose(new_c, ct, atol=1e-5)


def test_attention_pool():
    torch.manual_seed(3)
    ctx = torch.randn(2, 7, 5)
    logits = torch.randn(2, 7)
    alpha, pooled = F.attention_pool(ctx, logits)
    assert torch.allclose(alpha.sum(1), torch.ones(2), atol=1e-6)
    ref = torch.einsum('bl,bld->bd', alpha, ctx)
    assert torch.allclose(pooled, ref, atol=1e-6)


def test_embedding():
    table = torch.randn(10, 4)
    ids = torch.tensor([1, 3, 3, 9])
    out = F.embedding(ids, table)
    assert torch.allclose(out[1], out[2])
    assert torch.allclose(out[0], table[1])


def test_masked_ce():
    torch.manual_seed(4)
    logits = torch.randn(3, 6)
    labels = torch.tensor([1, 2, 0])
    mask = torch.tensor([1.0, 0.0, 1.0])
    ce = F.masked_softmax_ce(logits, labels, mask)
    assert ce[1].item() == 0.0
    ref = torch.nn.functional.cross_entropy(logits, labels,
                                            reduction='none') * mask
    assert torch.allclose(ce, ref)


def test_dropout_train_eval():
    x = torch.ones(1000)
    assert torch.equal(F.dropout(x, 0.5, training=False), x)
    y = F.dropout(x, 0.5, training=True)
    kept = (y > 0).float().mean().item()
    assert 0.3 < kept < 0.7
    assert np.isclose(y.max().item(), 2.0)  # inverted dropout scaling
