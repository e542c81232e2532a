#!/usr/bin/env python3
"""Per-shape wgrad: conv3x3_wgrad (in-tree) vs MIOpen (torch.nn.grad)."""
import sys
import torch
sys.path.insert(0, '.')
from sat_amd import _C

DEV = 'cuda'
B = 16  # train_cnn config #4 per-GPU batch

SHAPES = [(64, 64, 224), (64, 128, 112), (128, 128, 112),
          (128, 256, 56), (256, 256, 56), (256, 512, 28),
          (512, 512, 28), (512, 512, 14)]


def timeit(fn, iters=20):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000


for Cin, Cout, H in SHAPES:
    x = torch.randn(B, Cin, H, H).to(DEV, torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    dy = torch.randn(B, Cout, H, H).to(DEV, torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    xpad = _C.pad1_nhwc(x)
    dy_rows = dy.permute(0, 2, 3, 1).reshape(B * H * H, Cout)

    dwf = _C.conv3x3_wgrad(xpad, dy_rows, H, H)
    got = dwf.reshape(Cout, 3, 3, Cin).permute(0, 3, 1, 2)
    ref = torch.nn.grad.conv2d_weight(
        x.float(), (Cout, Cin, 3, 3), dy.float(), padding=1)
    rel = ((got - ref).abs().max() / ref.abs().max().clamp_min(1e-6)) \
        .item()

    t_mine = timeit(lambda: _C.conv3x3_wgrad(xpad, dy_rows, H, H))
    wshape = (Cout, Cin, 3, 3)
    t_mi = timeit(lambda: torch.nn.grad.conv2d_weight(
        x, wshape, dy, padding=1))
    gf = 2.0 * B * H * H * Cout * Cin * 9 / 1e9
    print('Cin%4d Cout%4d H%4d  rel %.4f  mine %7.1fus (%5.0f TF)  '
          'miopen %7.1fus (%5.0f TF)' %
          (Cin, Cout, H, rel, t_mine, gf / t_mine * 1e6,
           t_mi, gf / t_mi * 1e6))
