"""NN primitive layer — the single funnel through which every learnable op
is created (parity with reference `utils/nn.py`).

Centralizes:
  * initializer policy — fc kernels uniform ±fc_kernel_initializer_scale
    (nn.py:29-31), conv kernels Xavier (nn.py:15);
  * regularization policy — L2 on fc/conv kernels at the configured scales,
    collected into `reg_loss()` exactly like TF's graph-level
    get_regularization_loss (nn.py:17-43, model.py:328);
  * dropout rates (fc_drop_rate, nn.py:107-114);
  * the freeze-CNN policy — conv/bn parameters are trainable only when
    `config.train_cnn` (nn.py:66), fc parameters only when `is_train`.

Modules created here call into sat_amd.ops, which dispatches to the CDNA4
HIP kernels on GPU and plain PyTorch on CPU.
"""

import math

import torch
import torch.nn as tnn

from .. import ops


class NN(object):
    """Policy object handed to every layer factory."""

    def __init__(self, config):
        self.config = config
        self.is_train = getattr(config, 'phase', 'train') == 'train'
        self.train_cnn = self.is_train and getattr(config, 'train_cnn', False)
        self.fc_scale = config.fc_kernel_initializer_scale
        self.fc_reg = config.fc_kernel_regularizer_scale \
            if self.is_train else 0.0
        self.conv_reg = config.conv_kernel_regularizer_scale \
            if self.train_cnn else 0.0
        self.fc_drop_rate = config.fc_drop_rate
        self.lstm_drop_rate = config.lstm_drop_rate
        # L1 activity regularizers (reference nn.py:23-27,39-43): applied
        # to the output of every activation-carrying fc/conv layer, gated
        # on is_train (fc) / train_cnn (conv) exactly like the kernels.
        self.fc_act_reg = getattr(
            config, 'fc_activity_regularizer_scale', 0.0) \
            if self.is_train else 0.0
        self.conv_act_reg = getattr(
            config, 'conv_activity_regularizer_scale', 0.0) \
            if self.train_cnn else 0.0
        # (param, scale) pairs for reg_loss; filled as layers are built
        self._regularized = []
        # per-forward activity-loss terms (cleared by reg_loss)
        self._act_losses = []

    def init_fc_(self, t):
        tnn.init.uniform_(t, -self.fc_scale, self.fc_scale)

    def init_conv_(self, t):
        tnn.init.xavier_uniform_(t)

    def register_fc_kernel(self, p):
        if self.fc_reg > 0:
            self._regularized.append((p, self.fc_reg))

    def register_conv_kernel(self, p):
        if self.conv_reg > 0:
            self._regularized.append((p, self.conv_reg))

    def add_activity_loss(self, y, scale):
        """Record scale · Σ|y| for one layer output (TF l1_regularizer
        as activity_regularizer, reference nn.py:23-27,39-43)."""
        if scale > 0 and torch.is_grad_enabled():
            self._act_losses.append(scale * y.float().abs().sum())

    def reg_loss(self):
        """Σ scale · l2_loss(w) (l2_loss = sum(w²)/2, TF semantics) plus
        any activity-regularizer terms recorded this forward.

        On GPU the kernel-L2 term runs as ONE multi-tensor foreach pass
        (a per-param eager loop cost ~55 launches/step)."""
        total = torch.zeros(())
        if self._regularized:
            if self._regularized[0][0].is_cuda:
                params = tuple(p for p, _ in self._regularized)
                scales = tuple(s for _, s in self._regularized)
                total = _RegL2.apply(scales, *params)
            else:
                total = sum(s * 0.5 * (p.float() ** 2).sum()
                            for p, s in self._regularized)
        if self._act_losses:
            act = sum(self._act_losses)
            self._act_losses = []
            total = total.to(act.device) + act
        return total

    def dropout(self, x):
        return ops.dropout(x, self.fc_drop_rate, self.is_train)


class _RegL2(torch.autograd.Function):
    """Multi-tensor Σ scale·0.5·||p||² with closed-form backward
    (d/dp = d·scale·p): ~6 launches forward + one foreach backward
    instead of per-param pow/reduce/mul/add chains."""

    _scale_cache = {}

    @staticmethod
    def forward(ctx, scales, *params):
        ctx.save_for_backward(*params)
        ctx.scales = scales
        norms = torch._foreach_norm(params, 2)
        sv = torch.stack([n.float() for n in norms])
        # cached device constant: creating it per call would be a host
        # copy inside the captured graph
        key = (scales, sv.device)
        sc = _RegL2._scale_cache.get(key)
        if sc is None:
            sc = torch.tensor(scales, dtype=torch.float32,
                              device=sv.device)
            _RegL2._scale_cache[key] = sc
        return (sv * sv * sc * 0.5).sum()

    @staticmethod
    def backward(ctx, d):
        params = ctx.saved_tensors
        gs = torch._foreach_mul(params, list(ctx.scales))
        gs = torch._foreach_mul(gs, d)
        return (None, *gs)


class Dense(tnn.Module):
    """tf.layers.dense analog (reference nn.py:85-105); kernel stored
    [out, in] (torch layout) for transpose-free MFMA consumption."""

    def __init__(self, nn_policy, in_dim, units, activation='tanh',
                 use_bias=True):
        super().__init__()
        self.activation = activation
        self._nn = nn_policy
        self.weight = tnn.Parameter(torch.empty(units, in_dim))
        nn_policy.init_fc_(self.weight)
        nn_policy.register_fc_kernel(self.weight)
        if use_bias:
            self.bias = tnn.Parameter(torch.zeros(units))
        else:
            self.register_parameter('bias', None)
        self._wc = None  # per-forward cast cache (set by precast())
        self._bc = None

    def precast(self, dtype):
        """Cast weights once per model forward; the cast stays in the
        autograd graph, so T uses of the layer accumulate into one fp32
        parameter grad (like TF's reuse_variables, model.py:312)."""
        self._wc = self.weight.to(dtype)
        self._bc = self.bias.to(dtype) if self.bias is not None else None

    def clear_cast(self):
        self._wc = None
        self._bc = None

    def forward(self, x):
        if self._wc is not None and self._wc.dtype == x.dtype:
            y = ops.dense(x, self._wc, self._bc, self.activation)
        else:
            w = self.weight.to(x.dtype)
            b = self.bias.to(x.dtype) if self.bias is not None else None
            y = ops.dense(x, w, b, self.activation)
        # activity regularizer only on activation-carrying layers
        # (reference nn.py:92-95)
        if self.activation is not None and self._nn.fc_act_reg > 0:
            self._nn.add_activity_loss(y, self._nn.fc_act_reg)
        return y


class Conv2d(tnn.Module):
    """conv2d + bias + optional ReLU (reference nn.py:45-70 defaults:
    3x3, stride 1, SAME padding, ReLU)."""

    def __init__(self, nn_policy, in_ch, out_ch, kernel_size=3, stride=1,
                 activation='relu', use_bias=True):
        super().__init__()
        self.stride = stride
        self.kernel_size = kernel_size
        self.activation = activation
        self._nn = nn_policy
        self._glds_conv = getattr(nn_policy.config, 'use_glds_conv',
                                  True)
        self.weight = tnn.Parameter(
            torch.empty(out_ch, in_ch, kernel_size, kernel_size))
        nn_policy.init_conv_(self.weight)
        nn_policy.register_conv_kernel(self.weight)
        if use_bias:
            self.bias = tnn.Parameter(torch.zeros(out_ch))
        else:
            self.register_parameter('bias', None)

    def forward(self, x):
        w = self.weight.to(x.dtype)
        b = self.bias.to(x.dtype) if self.bias is not None else None
        k, st = self.kernel_size, self.stride
        # chained-pad handshake: a producing conv may hand over a 1-px
        # zero-bordered output (tagged `_sat_pad`); a 3x3/s1 consumer
        # marked _accept_padded uses it directly, anyone else unpads
        pad_hw = getattr(x, '_sat_pad', None)
        if pad_hw is not None and not (
                getattr(self, '_accept_padded', False)
                and k == 3 and st == 1
                and not torch.is_grad_enabled()):
            x = x[:, :, 1:-1, 1:-1] \
                .contiguous(memory_format=torch.channels_last)
            pad_hw = None
        # --train_cnn: hand-written forward/dgrad/wgrad autograd triple
        # (reference model.py:505-511 backward surface on in-tree kernels)
        if (torch.is_grad_enabled()
                and (x.requires_grad or w.requires_grad)
                and self._glds_conv and k == 3 and st == 1
                and self.activation in ('relu', None)):
            from ..ops.convgrad import Conv3x3Train, conv3x3_train_ok
            if conv3x3_train_ok(x, w):
                y = Conv3x3Train.apply(x, w, b,
                                       self.activation == 'relu')
                if self.activation is not None \
                        and self._nn.conv_act_reg > 0:
                    self._nn.add_activity_loss(y, self._nn.conv_act_reg)
                return y
        if (x.is_cuda and x.dtype == torch.bfloat16
                and not torch.is_grad_enabled()
                and k == 3 and st == 1 and w.shape[1] == 3
                and w.shape[0] % 8 == 0 and w.shape[0] <= 128
                and x.is_contiguous(memory_format=torch.channels_last)):
            # 3-channel first layer: MIOpen NHWC bf16 falls back to a
            # 2.4 ms CK kernel here; the direct kernel is memory-bound
            from ..ops import hip
            if hip.available():
                from sat_amd import _C
                emit = bool(getattr(self, '_emit_padded', False)
                            and self._glds_conv)
                y = _C.conv3_fwd(
                    x, w.contiguous(),
                    b if b is not None
                    else torch.empty(0, dtype=x.dtype, device=x.device),
                    self.activation == 'relu', emit)
                if emit:
                    y._sat_pad = (x.shape[2], x.shape[3])
                return y
        # frozen GPU implicit-GEMM paths (per-shape winners,
        # profiles/r01_conv_shapes.txt + r02 8-phase kernel):
        #  * Cout%256==0 with enough output pixels to fill the chip at a
        #    256^2 tile: the 8-phase deep-pipelined igemm (conv8p.hip) —
        #    the Cin>=256 layers MIOpen used to keep;
        #  * Cin=64: register-staged MFMA conv (337 vs 407 us on conv1_2)
        #  * Cin<=128 & Cout>=128: the XOR-swizzled glds variant
        #    (120 vs 169 us on conv2_1); `use_glds_conv` gates both
        #    LDS-staged variants.
        if (x.is_cuda and x.dtype == torch.bfloat16
                and not torch.is_grad_enabled()
                and k == 3 and st == 1
                and w.shape[0] % 8 == 0 and w.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            from ..ops import hip
            if hip.available():
                from sat_amd import _C
                Cout, Cin = w.shape[0], w.shape[1]
                H = pad_hw[0] if pad_hw else x.shape[2]
                W = pad_hw[1] if pad_hw else x.shape[3]
                M = x.shape[0] * H * W
                relu = self.activation == 'relu'
                # 8p wins every Cout%256 shape with M>=12544; at
                # conv5-batch-32 (M=6272) MIOpen gk still wins 77 vs
                # 99us (profiles/r02_conv_shapes.txt) — evidence-routed
                use_8p = (self._glds_conv and Cout % 256 == 0
                          and Cin % 64 == 0 and M >= 12544)
                use_glds = (self._glds_conv and not use_8p
                            and Cout >= 128 and Cout % 8 == 0
                            and Cin % 64 == 0 and Cin <= 128)
                # Cout=64 class (conv1_2): glds64 525 TF vs the register
                # igemm's 350 (profiles/r02_conv_shapes.txt)
                use_g64 = (self._glds_conv and not use_8p
                           and not use_glds and Cout < 128
                           and Cin % 64 == 0)
                use_ig64 = (not use_8p and not use_glds and not use_g64
                            and Cin == 64 and pad_hw is None)
                if use_8p or use_glds or use_g64 or use_ig64:
                    if getattr(self, '_w_ohwi', None) is None or \
                            self._w_ohwi_ver != self.weight._version:
                        self._w_ohwi_ver = self.weight._version
                        self._w_ohwi = w.permute(0, 2, 3, 1).contiguous() \
                            .reshape(w.shape[0], -1)
                    eb = (b if b is not None else
                          torch.empty(0, dtype=x.dtype, device=x.device))
                    if use_ig64:
                        return _C.conv_igemm_fwd(x, self._w_ohwi, eb,
                                                 relu)
                    xp = x if pad_hw is not None else _C.pad1_nhwc(x)
                    if use_8p:
                        return _C.conv_igemm_8p_fwd(
                            xp, self._w_ohwi, eb, H, W, relu)
                    if use_glds:
                        return _C.conv_igemm_glds_fwd(
                            xp, self._w_ohwi, eb, H, W, relu)
                    return _C.conv_igemm_glds64_fwd(
                        xp, self._w_ohwi, eb, H, W, relu)
        # 1x1/s1 conv IS a GEMM over the NHWC row view: route through
        # the in-tree tiled MFMA GEMM instead of MIOpen (ResNet50's
        # bottleneck convs; bias handled by the GEMM epilogue)
        if (x.is_cuda and x.dtype == torch.bfloat16
                and not torch.is_grad_enabled()
                and k == 1 and st == 1 and pad_hw is None
                and self._glds_conv
                and w.shape[0] % 8 == 0 and w.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            from ..ops import hip
            if hip.available():
                from sat_amd import _C
                B_, Cin, H, W = x.shape
                Cout = w.shape[0]
                rows = x.permute(0, 2, 3, 1).reshape(B_ * H * W, Cin)
                if getattr(self, '_w_1x1', None) is None or \
                        self._w_1x1_ver != self.weight._version:
                    self._w_1x1_ver = self.weight._version
                    self._w_1x1 = w.reshape(Cout, Cin).contiguous()
                eb = (b if b is not None else
                      torch.empty(0, dtype=x.dtype, device=x.device))
                act = 2 if self.activation == 'relu' else 0
                y = _C.dense_fwd(rows, self._w_1x1, eb, act)
                return y.reshape(B_, H, W, Cout).permute(0, 3, 1, 2)
        if pad_hw is not None:
            # padded input fell through to a library path: unpad first
            x = x[:, :, 1:-1, 1:-1] \
                .contiguous(memory_format=torch.channels_last)
        # frozen GPU path: fused NHWC bias+ReLU kernel after the MIOpen
        # conv instead of two separate eager elementwise passes
        fuse_epi = (x.is_cuda and x.dtype == torch.bfloat16
                    and not torch.is_grad_enabled() and b is not None
                    and self.weight.shape[0] % 8 == 0)
        conv_bias = None if fuse_epi else b
        ph = _same_pad(x.shape[2], k, st)
        pw = _same_pad(x.shape[3], k, st)
        if ph % 2 == 0 and pw % 2 == 0:
            # symmetric SAME: no pad-copy kernel, pad inside the conv
            y = torch.nn.functional.conv2d(x, w, conv_bias, stride=st,
                                           padding=(ph // 2, pw // 2))
        else:
            x = torch.nn.functional.pad(
                x, (pw // 2, pw - pw // 2, ph // 2, ph - ph // 2))
            y = torch.nn.functional.conv2d(x, w, conv_bias, stride=st)
        if fuse_epi and y.is_contiguous(
                memory_format=torch.channels_last):
            from ..ops import hip
            if hip.available():
                from sat_amd import _C
                _C.bias_act_nhwc(y, b, self.activation == 'relu')
                return y
            y = y + b.reshape(1, -1, 1, 1)
        elif fuse_epi:
            y = y + b.reshape(1, -1, 1, 1)
        if self.activation == 'relu':
            y = torch.relu(y)
        # conv activity regularizer (reference nn.py:54-57): only on
        # activation-carrying layers, only when the CNN trains — which is
        # exactly when this autograd path (not the inference kernels) runs
        if self.activation is not None and self._nn.conv_act_reg > 0:
            self._nn.add_activity_loss(y, self._nn.conv_act_reg)
        return y


def _same_pad(size, k, s):
    out = math.ceil(size / s)
    return max((out - 1) * s + k - size, 0)


def _pad_same(x, k, s):
    """TF 'SAME' padding (asymmetric: extra on bottom/right)."""
    ph = _same_pad(x.shape[2], k, s)
    pw = _same_pad(x.shape[3], k, s)
    if ph == 0 and pw == 0:
        return x
    return torch.nn.functional.pad(
        x, (pw // 2, pw - pw // 2, ph // 2, ph - ph // 2))


class MaxPool2d(tnn.Module):
    """max_pool2d 2x2 s2 SAME (reference nn.py:72-83)."""

    def __init__(self, kernel_size=2, stride=2):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16
                and not torch.is_grad_enabled()
                and self.kernel_size == 2 and self.stride == 2
                and x.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            from ..ops import hip
            if hip.available():
                from sat_amd import _C
                return _C.maxpool2x2_nhwc(x)
        x = _pad_same(x, self.kernel_size, self.stride)
        return torch.nn.functional.max_pool2d(
            x, self.kernel_size, self.stride)


class BatchNorm(tnn.Module):
    """batch_norm gated on train_cnn (reference nn.py:116-125).

    Frozen GPU path: inference BN is a fixed per-channel scale/shift
    (scale = gamma/sqrt(var+eps), shift = beta - mean*scale), applied by
    ONE NHWC kernel with optional fused ReLU — instead of the eager
    fp32-cast + BN + bf16-cast chain (3 kernels x 53 BNs on ResNet50)."""

    def __init__(self, nn_policy, num_features):
        super().__init__()
        self.train_cnn = nn_policy.train_cnn
        self.bn = tnn.BatchNorm2d(num_features, eps=1e-3, momentum=0.01)
        self._ss = None  # cached (scale, shift) bf16 (frozen only)

    def forward(self, x, relu=False):
        if self.train_cnn:
            y = self.bn(x.float()).to(x.dtype)
            return torch.relu(y) if relu else y
        if (x.is_cuda and x.dtype == torch.bfloat16
                and not torch.is_grad_enabled()
                and x.shape[1] % 8 == 0
                and x.is_contiguous(memory_format=torch.channels_last)):
            from ..ops import hip
            if hip.available():
                from sat_amd import _C
                # all four stat/affine tensors key the cache: a partial
                # load_cnn/checkpoint match may in-place update only
                # running_mean or bias, which must invalidate too
                ver = (self.bn.running_var._version,
                       self.bn.running_mean._version,
                       self.bn.weight._version,
                       self.bn.bias._version)
                if (self._ss is None or self._ss[0].device != x.device
                        or getattr(self, '_ss_ver', None) != ver):
                    self._ss_ver = ver
                    bn = self.bn
                    scale = (bn.weight /
                             torch.sqrt(bn.running_var + bn.eps))
                    shift = bn.bias - bn.running_mean * scale
                    self._ss = (scale.to(x.device, torch.bfloat16)
                                .contiguous(),
                                shift.to(x.device, torch.bfloat16)
                                .contiguous())
                _C.scale_bias_act_nhwc(x, self._ss[0], self._ss[1], relu)
                return x
        training = self.bn.training
        self.bn.eval()
        y = self.bn(x.float()).to(x.dtype)
        if training:
            self.bn.train()
        return torch.relu(y) if relu else y
