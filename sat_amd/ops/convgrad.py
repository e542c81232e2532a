"""Trainable 3x3 conv on the in-tree CDNA4 kernels (--train_cnn path).

The reference trains the CNN through TF's autodiff (model.py:505-511);
here the whole conv triple is hand-written:
  * forward  — the same implicit-GEMM kernels the frozen path uses
    (conv8p.hip 8-phase / conv3.hip glds / register igemm);
  * dgrad    — the SAME forward kernels on the padded dy with the weight
    flipped and io-transposed (dx = conv3x3(dy, rot180(W)^T)), so the
    8-phase pipeline serves backward for free;
  * wgrad    — conv_bwd.hip: transpose-staged MFMA reduction over all
    output pixels with split-K fp32 atomics, emitting dW directly in the
    OHWI layout;
  * dbias    — column sum over the NHWC row view;
  * the ReLU epilogue is fused in forward and peeled in backward via the
    shared act_bwd kernel.

Applies to Cin % 64 == 0 layers (12 of VGG16's 13); conv1_1 (Cin=3)
stays on torch autograd — its wgrad K-tile is 27 wide, below the MFMA
staging grain, and it is <2% of CNN backward time.
"""

import torch

ACT_RELU = 2


def _route_fwd(x, w_ohwi, eb, relu):
    """Shape-routed igemm forward (same policy as models.nn.Conv2d)."""
    from sat_amd import _C
    Cout = w_ohwi.shape[0]
    Cin = x.shape[1]
    H, W = x.shape[2], x.shape[3]
    M = x.shape[0] * H * W
    if Cout % 256 == 0 and Cin % 64 == 0 and M >= 3136:
        return _C.conv_igemm_8p_fwd(_C.pad1_nhwc(x), w_ohwi, eb, H, W,
                                    relu)
    if Cout >= 128 and Cout % 8 == 0 and Cin % 64 == 0 and Cin <= 128:
        return _C.conv_igemm_glds_fwd(_C.pad1_nhwc(x), w_ohwi, eb, H, W,
                                      relu)
    return _C.conv_igemm_fwd(x, w_ohwi, eb, relu)


class Conv3x3Train(torch.autograd.Function):
    """3x3/s1/SAME conv + bias + optional ReLU with hand-written
    forward/dgrad/wgrad (NHWC bf16, Cin % 64 == 0, Cout % 64 == 0)."""

    @staticmethod
    def forward(ctx, x, w, bias, relu):
        from sat_amd import _C
        w_ohwi = w.permute(0, 2, 3, 1).contiguous() \
            .reshape(w.shape[0], -1)
        eb = (bias if bias is not None else
              torch.empty(0, dtype=x.dtype, device=x.device))
        y = _route_fwd(x, w_ohwi, eb, relu)
        if relu:
            ctx.save_for_backward(x, w, y)  # y needed for the ReLU peel
        else:
            ctx.save_for_backward(x, w)
        ctx.relu = relu
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        from sat_amd import _C
        if ctx.relu:
            x, w, y = ctx.saved_tensors
        else:
            x, w = ctx.saved_tensors
            y = None
        Cout, Cin = w.shape[0], w.shape[1]
        B, _, H, W = x.shape
        M = B * H * W

        dy = dy.contiguous(memory_format=torch.channels_last)
        # NHWC storage viewed as [M, Cout] rows (contiguous, zero-copy)
        dy_rows = dy.permute(0, 2, 3, 1).reshape(M, Cout)
        if not dy_rows.is_contiguous():
            dy_rows = dy_rows.contiguous()
        if ctx.relu:
            y_rows = y.permute(0, 2, 3, 1).reshape(M, Cout)
            dy_rows = _C.act_bwd(dy_rows, y_rows, ACT_RELU)
            # back to a channels_last 4D view for dgrad
            dy = dy_rows.reshape(B, H, W, Cout).permute(0, 3, 1, 2)

        dbias = dy_rows.float().sum(0).to(dy.dtype) \
            if ctx.has_bias else None

        xpad = _C.pad1_nhwc(x)
        dwf = _C.conv3x3_wgrad(xpad, dy_rows, H, W)     # [Cout, 9*Cin]
        # [Cout,3,3,Cin] IS the channels_last storage order of the OIHW
        # weight — match w's layout so the fused Adam's flat-storage
        # iteration pairs p/g/m/v correctly
        if w.is_contiguous(memory_format=torch.channels_last):
            dw = dwf.reshape(Cout, 3, 3, Cin).to(w.dtype) \
                .permute(0, 3, 1, 2)
        else:
            dw = dwf.reshape(Cout, 3, 3, Cin).permute(0, 3, 1, 2) \
                .contiguous().to(w.dtype)

        dx = None
        if ctx.needs_input_grad[0]:
            # dx = conv3x3(dy, rot180(W) io-transposed), OHWI repack
            wd = w.flip(2, 3).permute(1, 2, 3, 0).reshape(Cin, -1) \
                .contiguous()
            eb = torch.empty(0, dtype=dy.dtype, device=dy.device)
            dx = _route_fwd(dy, wd, eb, False)

        return dx, dw, dbias, None


def conv3x3_train_ok(x, w, config=None):
    """Eligibility for the hand-written training conv."""
    from . import hip
    return (x.is_cuda and x.dtype == torch.bfloat16
            and w.shape[2] == 3 and w.shape[3] == 3
            and w.shape[0] % 64 == 0 and w.shape[1] % 64 == 0
            and x.is_contiguous(memory_format=torch.channels_last)
            and hip.available())
