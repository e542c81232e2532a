"""Reference (plain PyTorch) implementations of the op layer.

These are the CPU path and the numerics ground truth for the HIP kernels.
Semantics mirror the TF ops the reference launches:

  * dense           — tf.layers.dense semantics; kernels are stored [out,in]
                      (torch Linear layout) so the GPU MFMA kernel reads a
                      [N,K]-contiguous B operand with no transpose
  * lstm_cell       — tf.nn.rnn_cell.LSTMCell (gate order i,j,f,o; forget
                      bias added pre-sigmoid; no peepholes)
  * attention_pool  — softmax over locations + Σ_l α_l·ctx_l
                      (reference model.py:435, :263-264)
  * embedding       — tf.nn.embedding_lookup (model.py:273)
  * masked_softmax_ce — tf.nn.sparse_softmax_cross_entropy_with_logits ×
                      mask (model.py:294-297)
"""

import torch
import torch.nn.functional as tF


def dense(x, weight, bias=None, activation=None):
    """act(x @ weight.T + bias); weight: [out, in]."""
    y = x.matmul(weight.t())
    if bias is not None:
        y = y + bias
    if activation == 'tanh':
        y = torch.tanh(y)
    elif activation == 'relu':
        y = torch.relu(y)
    elif activation not in (None, 'none'):
        raise ValueError('unknown activation %r' % (activation,))
    return y


def lstm_cell(x, h, c, weight, bias, forget_bias=1.0):
    """TF LSTMCell step.

    x: [B, I], h/c: [B, H], weight: [4H, I+H] with gate order (i, j, f, o),
    bias: [4H].  Returns (new_h, new_c).
    """
    gates = torch.cat([x, h], dim=1).matmul(weight.t()) + bias
    i, j, f, o = gates.chunk(4, dim=1)
    new_c = c * torch.sigmoid(f + forget_bias) + \
        torch.sigmoid(i) * torch.tanh(j)
    new_h = torch.tanh(new_c) * torch.sigmoid(o)
    return new_h, new_c


def attention_pool(contexts, logits):
    """contexts: [B, L, D], logits: [B, L] -> (alpha [B,L], context [B,D])."""
    alpha = torch.softmax(logits, dim=1)
    context = (contexts * alpha.unsqueeze(2)).sum(dim=1)
    return alpha, context


def attention_tail(t1, t2, v, contexts, p, training):
    """Attention tail (reference model.py:425-435 + :263-264):
    t = dropout(t1 + tiled t2); logits = t @ v; alpha = softmax over L;
    context = Σ_l α_l·ctx_l.  Returns (alpha [B,L], context [B,D])."""
    B, L = contexts.shape[0], contexts.shape[1]
    t = t1 + t2.repeat_interleave(L, dim=0)
    t = dropout(t, p, training)
    logits = t.matmul(v).reshape(B, L)
    return attention_pool(contexts, logits)


def embedding(ids, table):
    return table[ids]


def masked_softmax_ce(logits, labels, mask):
    """logits: [B, V], labels: [B] int64, mask: [B] -> masked CE [B]."""
    ce = tF.cross_entropy(logits.float(), labels, reduction='none')
    return ce * mask


def dropout(x, rate, training):
    if not training or rate <= 0.0:
        return x
    return tF.dropout(x, p=rate, training=True)
