"""CNN encoders: VGG16 and ResNet50 conv stacks emitting a spatial feature
grid (parity with reference `model.py:24-188`).

VGG16 (model.py:24-60): 13 conv3x3-s1-same+ReLU layers in 5 blocks with 4
max-pools between the first 4 blocks; output conv5_3 is [B,14,14,512],
reshaped to a [B,196,512] context grid.

ResNet50 (model.py:62-188): conv1 7x7-s2 + BN + ReLU + maxpool 3x3-s2, then
bottleneck stages (3,4,6,3) with 1-3-1 convs, BN everywhere, projection
shortcuts at stage entry; output is [B,7,7,2048] -> [B,49,2048].

Conv compute is `sat_amd.models.nn.Conv2d`, which routes per shape to
the in-tree CDNA4 kernels (direct 3-ch conv, glds/glds64/8-phase
implicit-GEMM, 1x1-as-GEMM; hand-written dgrad/wgrad under
--train_cnn) with MIOpen only where it still measures faster
(profiles/r02_conv_shapes.txt).  Frozen-CNN policy: parameters are
created non-trainable unless config.train_cnn (reference nn.py:66).
"""

import torch
import torch.nn as tnn

from .nn import BatchNorm, Conv2d, MaxPool2d


class VGG16(tnn.Module):
    # (name, out_ch) per conv layer; '|' marks a pool
    _PLAN = [
        ('conv1_1', 64), ('conv1_2', 64), 'pool',
        ('conv2_1', 128), ('conv2_2', 128), 'pool',
        ('conv3_1', 256), ('conv3_2', 256), ('conv3_3', 256), 'pool',
        ('conv4_1', 512), ('conv4_2', 512), ('conv4_3', 512), 'pool',
        ('conv5_1', 512), ('conv5_2', 512), ('conv5_3', 512),
    ]

    num_ctx = 196
    dim_ctx = 512

    def __init__(self, nn_policy):
        super().__init__()
        layers = []
        in_ch = 3
        for item in self._PLAN:
            if item == 'pool':
                layers.append(MaxPool2d(2, 2))
            else:
                name, out_ch = item
                conv = Conv2d(nn_policy, in_ch, out_ch, 3, 1, 'relu')
                self.add_module(name, conv)
                layers.append(conv)
                in_ch = out_ch
        self._layers = layers
        # chained-pad fast path: conv1_1's direct kernel emits a 1-px
        # zero-bordered output that conv1_2's glds igemm consumes
        # without a pad pass (~70 us/step at 224^2 — models/nn.py)
        self.conv1_1._emit_padded = True
        self.conv1_2._accept_padded = True

    def forward(self, images):
        """images: [B,3,224,224] -> contexts [B,196,512]."""
        x = images
        for layer in self._layers:
            x = layer(x)
        # [B,512,14,14] -> [B,196,512]; for channels_last (GPU) the NHWC
        # physical layout makes this permute+reshape a zero-copy view
        return x.permute(0, 2, 3, 1).reshape(
            x.shape[0], -1, x.shape[1]).contiguous()


class _Bottleneck(tnn.Module):
    def __init__(self, nn_policy, in_ch, mid_ch, stride, project):
        super().__init__()
        out_ch = mid_ch * 4
        self.project = project
        if project:
            self.shortcut = Conv2d(nn_policy, in_ch, out_ch, 1, stride,
                                   None, use_bias=False)
            self.shortcut_bn = BatchNorm(nn_policy, out_ch)
        self.conv_a = Conv2d(nn_policy, in_ch, mid_ch, 1, stride, None,
                             use_bias=False)
        self.bn_a = BatchNorm(nn_policy, mid_ch)
        self.conv_b = Conv2d(nn_policy, mid_ch, mid_ch, 3, 1, None,
                             use_bias=False)
        self.bn_b = BatchNorm(nn_policy, mid_ch)
        self.conv_c = Conv2d(nn_policy, mid_ch, out_ch, 1, 1, None,
                             use_bias=False)
        self.bn_c = BatchNorm(nn_policy, out_ch)

    def forward(self, x):
        if self.project:
            sc = self.shortcut_bn(self.shortcut(x))
        else:
            sc = x
        y = self.bn_a(self.conv_a(x), relu=True)
        y = self.bn_b(self.conv_b(y), relu=True)
        y = self.bn_c(self.conv_c(y))
        return torch.relu(y + sc)


class ResNet50(tnn.Module):
    _STAGES = [(64, 3, 1), (128, 4, 2), (256, 6, 2), (512, 3, 2)]

    num_ctx = 49
    dim_ctx = 2048

    def __init__(self, nn_policy):
        super().__init__()
        self.conv1 = Conv2d(nn_policy, 3, 64, 7, 2, None, use_bias=False)
        self.bn1 = BatchNorm(nn_policy, 64)
        self.pool1 = MaxPool2d(3, 2)
        blocks = []
        in_ch = 64
        for s, (mid, n, stride) in enumerate(self._STAGES, start=2):
            for b in range(n):
                blk = _Bottleneck(nn_policy, in_ch, mid,
                                  stride if b == 0 else 1, b == 0)
                self.add_module('res%d%s' % (s, chr(ord('a') + b)), blk)
                blocks.append(blk)
                in_ch = mid * 4
        self._blocks = blocks

    def forward(self, images):
        """images: [B,3,224,224] -> contexts [B,49,2048]."""
        x = self.bn1(self.conv1(images), relu=True)
        x = self.pool1(x)
        for blk in self._blocks:
            x = blk(x)
        return x.permute(0, 2, 3, 1).reshape(
            x.shape[0], -1, x.shape[1]).contiguous()


def build_encoder(name, nn_policy):
    if name == 'vgg16':
        return VGG16(nn_policy)
    if name == 'resnet50':
        return ResNet50(nn_policy)
    raise ValueError('unknown cnn %r' % (name,))
