"""Hand-written BPTT for the attention-LSTM decoder (GPU bf16 path).

The reference gets its training step as one static TF graph executed by the
C++ runtime; autograd-per-op in a Python loop pays ~1500
kernel dispatches and ~300 gradient-accumulation adds per step instead.
This module runs the whole T-step decoder as ONE torch.autograd.Function:

  * forward: the 20 teacher-forced steps as direct _C kernel calls
    (dense MFMA GEMMs, fused attention tail, fused LSTM gates, fused CE),
    stashing exactly the tensors backward needs;
  * backward: the reverse-time loop over fused backward kernels with
    hand-carried recurrent gradients (output / state-h / cell), where
    - CE backward runs ONCE batched over [T·B, V],
    - the small-M (batch-sized) weight-gradient GEMMs run ONCE per weight
      batched over [T·B, ·] instead of T matmuls + T-1 accumulation adds,
    - the embedding scatter-add runs ONCE over [T·B] ids,
    - every dropout mask regenerates from the counter-based hash
      (sat_amd/ops/csrc/kernels.hip) — zero mask storage.

Semantics are identical to the per-op path (reference model.py:259-334):
same gate order, same dropout sites (input/output/state-h, fc layers),
same doubly-stochastic attention-loss wiring through masked alphas.

Supported shape: the reference default architecture (2-layer attend MLP,
2-layer decode MLP).  Other configs use the per-op autograd loop.
"""

import torch

from sat_amd import _C

ACT_NONE, ACT_TANH, ACT_RELU = 0, 1, 2


def _drop(x, seed, p, salt):
    if p <= 0.0:
        return x
    return _C.hash_dropout(x, seed, p, salt)


class DecoderBPTT(torch.autograd.Function):
    @staticmethod
    def forward(ctx_ag, contexts, init_memory, init_output, sentences,
                masks, emb, w1a, b1a, w1b, b1b, v, wl, bl, wd1, bd1,
                wd2, bd2, seed, p_fc, p_lstm, train_cnn):
        B, L, D = contexts.shape
        T = sentences.shape[1]
        A = w1a.shape[0]
        H = init_memory.shape[1]
        E = emb.shape[1]
        V = wd2.shape[0]
        Dd = wd1.shape[0]
        I = D + E
        dev = contexts.device

        ctx_flat = contexts.reshape(B * L, D)
        # all T context-dropout slabs in one kernel (salt t*16+0); the
        # buffer also feeds the batched dW_1a GEMM in backward.  fc_1a
        # (the attention context projection) does not depend on the
        # recurrence, so ALL T steps run as ONE batched MFMA GEMM here
        # instead of 20 per-step launches.
        if p_fc > 0.0:
            CDROP = _C.hash_dropout_steps(ctx_flat, seed, p_fc, 0, 16, T)
            T1 = _C.dense_fwd(CDROP.reshape(T * B * L, D), w1a, b1a, 1)
        else:
            CDROP = None
            T1 = _C.dense_fwd(ctx_flat, w1a, b1a, 1)  # shared by all t

        # forward-side batched buffers (consumed by backward's batched dW)
        XH = torch.empty(T * B, I + H, dtype=torch.bfloat16, device=dev)
        EXPD = torch.empty(T * B, H + D + E, dtype=torch.bfloat16,
                           device=dev)
        HD = torch.empty(T * B, Dd, dtype=torch.bfloat16, device=dev)
        HID = torch.empty(T * B, Dd, dtype=torch.bfloat16, device=dev)
        ODROP = torch.empty(T * B, H, dtype=torch.bfloat16, device=dev)
        LOGITS = torch.empty(T * B, V, dtype=torch.bfloat16, device=dev)

        t1s, t2s, tdrops, alphas = [], [], [], []
        gates_l, cprev_l, hid_l = [], [], []
        preds = []

        labels_cat = sentences.t().reshape(-1)          # [T·B] step-major
        masks_cat = masks.t().reshape(-1).contiguous()  # [T·B] float

        memory = init_memory
        state_h = init_output
        last_word = torch.zeros(B, dtype=torch.int64, device=dev)
        empty_b = _EMPTY_B(dev)
        fuse_small = (B <= 128 and (I + H) % 32 == 0
                      and (H + D + E) % 32 == 0)

        # attend input for step 0 (later steps' come fused out of
        # expand_fuse at step t-1)
        _C.hash_dropout_out(init_output, seed, p_fc, 1, ODROP[0:B])

        for t in range(T):
            s = t * 16
            sl = slice(t * B, (t + 1) * B)

            t1 = T1.reshape(-1, A)[t * B * L:(t + 1) * B * L] \
                if CDROP is not None else T1
            t2 = _C.dense_fwd(ODROP[sl], w1b, b1b, ACT_TANH)
            tdrop, att_logits = _C.attn_scores_fused(
                t1, t2, v, seed, p_fc, s + 2, L)
            alpha, pooled = _C.attn_pool_fwd(contexts, att_logits)

            _C.lstm_in_fuse(pooled, emb, last_word, state_h, seed,
                            p_lstm, s + 3, XH[sl])
            if fuse_small:
                gates, h_raw, c_new = _C.dense_lstm_fwd(
                    XH[sl], wl, bl, memory, 1.0)
            else:
                gates = _C.dense_fwd(XH[sl], wl, bl, ACT_NONE)
                h_raw, c_new = _C.lstm_pointwise_fwd(gates, memory, 1.0)
            od_next = ODROP[(t + 1) * B:(t + 2) * B] if t + 1 < T \
                else empty_b
            out_t, sth_t = _C.expand_fuse(
                h_raw, pooled, emb, last_word, seed, EXPD[sl], od_next,
                p_lstm, p_fc, s)
            if fuse_small:
                _C.dense_drop_fwd(EXPD[sl], wd1, bd1, ACT_TANH, seed,
                                  p_fc, s + 7, HID[sl], HD[sl])
                hid = HID[sl]
            else:
                hid = _C.dense_fwd_out(EXPD[sl], wd1, bd1, ACT_TANH,
                                       HID[sl])
                _C.hash_dropout_out(hid, seed, p_fc, s + 7, HD[sl])
            _C.dense_fwd_out(HD[sl], wd2, bd2, ACT_NONE, LOGITS[sl])

            t1s.append(t1)
            t2s.append(t2)
            tdrops.append(tdrop)
            alphas.append(alpha)
            gates_l.append(gates)
            cprev_l.append(memory)

            memory = c_new
            state_h = sth_t
            last_word = labels_cat[sl]

        # batched loss / argmax / attention accumulation (one pass each)
        CE, LSE = _C.ce_fwd(LOGITS, labels_cat, masks_cat)
        ce = CE.reshape(T, B).t().contiguous()           # [B,T]
        predictions = LOGITS.reshape(T, B, V).argmax(dim=2) \
            .t().contiguous()                            # [B,T]
        alpha_stack = torch.stack(alphas)                # [T,B,L]
        attn_acc = (alpha_stack
                    * masks.t().reshape(T, B, 1)).sum(dim=0)

        ctx_ag.save_for_backward(
            contexts, emb, w1a, b1a, w1b, b1b, v, wl, bl, wd1, bd1,
            wd2, bd2, seed, XH, EXPD, HD, HID, ODROP, LOGITS, LSE,
            labels_cat, masks_cat, masks)
        ctx_ag.cdrop = CDROP
        ctx_ag.saved_lists = (t1s, t2s, tdrops, alphas, gates_l, cprev_l)
        ctx_ag.dims = (B, L, D, T, A, H, E, V, Dd, I)
        ctx_ag.p_fc = p_fc
        ctx_ag.p_lstm = p_lstm
        ctx_ag.train_cnn = train_cnn
        ctx_ag.mark_non_differentiable(predictions)
        return ce, attn_acc, predictions

    @staticmethod
    def backward(ctx_ag, d_ce, d_attn, _d_pred):
        (contexts, emb, w1a, b1a, w1b, b1b, v, wl, bl, wd1, bd1,
         wd2, bd2, seed, XH, EXPD, HD, HID, ODROP, LOGITS, LSE,
         labels_cat, masks_cat, masks) = ctx_ag.saved_tensors
        (t1s, t2s, tdrops, alphas, gates_l,
         cprev_l) = ctx_ag.saved_lists
        B, L, D, T, A, H, E, V, Dd, I = ctx_ag.dims
        p_fc = ctx_ag.p_fc
        p_lstm = ctx_ag.p_lstm
        dev = contexts.device
        need_dctx = ctx_ag.train_cnn and ctx_ag.needs_input_grad[0]

        ctx_flat = contexts.reshape(B * L, D)

        # ---- batched CE backward over [T·B, V] ----
        dce_cat = d_ce.t().reshape(-1).contiguous().float()
        DL = _C.ce_bwd(LOGITS, labels_cat, masks_cat, LSE, dce_cat)

        # ---- batched decode-MLP input grads (no recurrence involved) ----
        DHD = DL.matmul(wd2)                      # [T·B, Dd]
        DHID = _C.hash_dropout_slabs(DHD, seed, p_fc, 7, 16, T)
        DP1 = _C.act_bwd(DHID, HID, ACT_TANH)     # dpre of dec fc_1
        DEXPD = DP1.matmul(wd1)                   # [T·B, H+D+E]

        # transposed weights for the per-step skinny GEMMs (x @ W forms)
        wl_t = wl.t().contiguous()
        w1b_t = w1b.t().contiguous()

        DG = torch.empty(T * B, 4 * H, dtype=torch.bfloat16, device=dev)
        DEMB = torch.empty(T * B, E, dtype=torch.bfloat16, device=dev)
        DPRE1B = torch.empty(T * B, A, dtype=torch.bfloat16, device=dev)
        DPRE1A = torch.empty(T * B * L, A, dtype=torch.bfloat16,
                             device=dev)
        CDROP = ctx_ag.cdrop

        dv_acc = torch.zeros_like(v, dtype=torch.float32)
        dctx_acc = torch.zeros_like(contexts) if need_dctx else None

        d_out_carry = torch.zeros(B, H, dtype=torch.bfloat16, device=dev)
        d_sth_carry = torch.zeros(B, H, dtype=torch.bfloat16, device=dev)
        dc_carry = torch.zeros(B, H, dtype=torch.bfloat16, device=dev)

        for t in range(T - 1, -1, -1):
            s = t * 16
            sl = slice(t * B, (t + 1) * B)
            dh_raw, dpool_dec, demb_dec = _C.dexp_fuse(
                DEXPD[sl], d_out_carry, d_sth_carry, seed, p_fc, p_lstm,
                s, D, E)
            dgates, dc_prev = _C.lstm_pointwise_bwd_out(
                gates_l[t], cprev_l[t], dh_raw, dc_carry, 1.0, DG[sl])
            dxh = _C.dense_fwd(dgates, wl_t, _EMPTY_B(dev), ACT_NONE)
            dpooled, d_sth_carry = _C.dx_fuse(
                dxh, dpool_dec, demb_dec, seed, DEMB[sl], p_lstm,
                s + 3, H)
            dc_carry = dc_prev

            dalpha_t = d_attn * masks[:, t].unsqueeze(1)
            dlog_att, dctx_t = _C.attn_pool_bwd(
                contexts, alphas[t], dalpha_t.contiguous(),
                dpooled, need_dctx)
            if need_dctx:
                dctx_acc += dctx_t
            dt1, dt2f, _dv = _C.attn_scores_bwd_acc(
                tdrops[t], v, dlog_att, seed, p_fc, s + 2, L, dv_acc)

            dpre1b = _C.act_bwd(dt2f.to(torch.bfloat16), t2s[t], ACT_TANH)
            DPRE1B[sl] = dpre1b
            dodrop = _C.dense_fwd(dpre1b, w1b_t, _EMPTY_B(dev), ACT_NONE)
            d_out_carry = _drop(dodrop, seed, p_fc, s + 1)

            sl_a = slice(t * B * L, (t + 1) * B * L)
            _C.act_bwd_out(dt1, t1s[t], ACT_TANH, DPRE1A[sl_a])
            if need_dctx:
                dcd = DPRE1A[sl_a].matmul(w1a)
                dctx_acc += _drop(dcd, seed, p_fc, s + 0) \
                    .reshape(B, L, D)

        # ---- batched weight grads ----
        if CDROP is not None:
            # hipBLASLt schedules this K=T*B*L GEMM on ~32 blocks (163 TF
            # measured); as a T-chunk bmm + sum it fills the chip
            dW1a = torch.bmm(
                DPRE1A.reshape(T, B * L, A).transpose(1, 2),
                CDROP.reshape(T, B * L, D)).sum(dim=0)
        else:
            # p_fc == 0: every step saw the same ctx_flat
            dW1a = DPRE1A.reshape(T, B * L, A).sum(0).t() \
                .matmul(ctx_flat)
        db1a = DPRE1A.float().sum(0)
        dWl = DG.t().matmul(XH)
        dbl = DG.float().sum(0)
        dWd2 = DL.t().matmul(HD)
        dbd2 = DL.float().sum(0)
        dWd1 = DP1.t().matmul(EXPD)
        dbd1 = DP1.float().sum(0)
        dW1b = DPRE1B.t().matmul(ODROP)
        db1b = DPRE1B.float().sum(0)
        demb_table = _C.embedding_bwd(
            torch.cat([torch.zeros(B, dtype=torch.int64, device=dev),
                       labels_cat[:-B]]),
            DEMB, emb.shape[0])

        d_init_output = (d_out_carry.float()
                         + d_sth_carry.float()).to(torch.bfloat16)
        d_init_memory = dc_carry

        bf = torch.bfloat16
        return (dctx_acc, d_init_memory, d_init_output, None, None,
                demb_table.to(bf), dW1a.to(bf), db1a.to(bf),
                dW1b.to(bf), db1b.to(bf), dv_acc.to(bf),
                dWl.to(bf), dbl.to(bf), dWd1.to(bf), dbd1.to(bf),
                dWd2.to(bf), dbd2.to(bf), None, None, None, None)


_EMPTY = {}


def _EMPTY_B(device):
    key = str(device)
    if key not in _EMPTY:
        _EMPTY[key] = torch.empty(0, device=device, dtype=torch.bfloat16)
    return _EMPTY[key]


def run_decoder_bptt(decoder, contexts, init_memory, init_output,
                     sentences, masks):
    """Run the fused BPTT decoder loop. Returns (ce [B,T], attentions
    [B,L], predictions [B,T])."""
    d = decoder
    return DecoderBPTT.apply(
        contexts, init_memory, init_output, sentences, masks,
        d._emb_c, d.att_fc_1a._wc, d.att_fc_1a._bc,
        d.att_fc_1b._wc, d.att_fc_1b._bc, d._att_vc,
        d._lstm_wc, d._lstm_bc,
        d.dec_fc_1._wc, d.dec_fc_1._bc, d.dec_fc_2._wc, d.dec_fc_2._bc,
        d._rng, d.nn.fc_drop_rate, d.nn.lstm_drop_rate,
        d.nn.train_cnn)
