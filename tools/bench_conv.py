import sys
import torch
sys.path.insert(0, '.')
from sat_amd import _C

DEV = 'cuda'
torch.backends.cudnn.benchmark = True

SHAPES = [  # (Cin, Cout, H) VGG16 layers 2..13 at batch 32
    (64, 64, 224), (64, 128, 112), (128, 128, 112),
    (128, 256, 56), (256, 256, 56),
    (256, 512, 28), (512, 512, 28),
    (512, 512, 14),
]
B = 32


def timeit(fn, iters=30):
    for _ in range(8):
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
    w = (torch.randn(Cout, Cin, 3, 3) * 0.02).to(DEV, torch.bfloat16)
    bias = torch.randn(Cout).to(DEV, torch.bfloat16)
    w_ohwi = w.permute(0, 2, 3, 1).contiguous().reshape(Cout, 9 * Cin)

    y = _C.conv_igemm_fwd(x, w_ohwi, bias, True)
    ref = torch.relu(torch.nn.functional.conv2d(
        x.float(), w.float(), bias.float(), padding=1))
    rel = ((y.float() - ref).abs().max() /
           ref.abs().max().clamp_min(1e-6)).item()

    t_mine = timeit(lambda: _C.conv_igemm_fwd(x, w_ohwi, bias, True))
    wl = w.contiguous(memory_format=torch.channels_last)
    t_mi = timeit(lambda: torch.relu(
        torch.nn.functional.conv2d(x, wl, bias, padding=1)))
    gf = 2.0 * B * H * H * Cout * Cin * 9 / 1e9
    print('Cin%4d Cout%4d H%4d  rel %.4f  mine %7.1fus (%5.0f TF)  '
          'miopen+relu %7.1fus (%5.0f TF)' %
          (Cin, Cout, H, rel, t_mine, gf / t_mine * 1e6,
           t_mi, gf / t_mi * 1e6))

print('== glds variant (padded input) ==')
for Cin, Cout, H in SHAPES:
    if Cout < 128:
        continue
    x = torch.randn(B, Cin, H, H).to(DEV, torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Cout, Cin, 3, 3) * 0.02).to(DEV, torch.bfloat16)
    bias = torch.randn(Cout).to(DEV, torch.bfloat16)
    w_ohwi = w.permute(0, 2, 3, 1).contiguous().reshape(Cout, 9 * Cin)
    xp = _C.pad1_nhwc(x)
    y = _C.conv_igemm_glds_fwd(xp, w_ohwi, bias, H, H, True)
    ref = torch.relu(torch.nn.functional.conv2d(
        x.float(), w.float(), bias.float(), padding=1))
    rel = ((y.float() - ref).abs().max() /
           ref.abs().max().clamp_min(1e-6)).item()
    t_pad = timeit(lambda: _C.pad1_nhwc(x))
    t_g = timeit(lambda: _C.conv_igemm_glds_fwd(xp, w_ohwi, bias, H, H,
                                                True))
    gf = 2.0 * B * H * H * Cout * Cin * 9 / 1e9
    print('Cin%4d Cout%4d H%4d  rel %.4f  glds %7.1fus (%5.0f TF) '
          '+pad %5.1fus' % (Cin, Cout, H, rel, t_g, gf / t_g * 1e6, t_pad))

print('== 8-phase pipelined variant (padded input, Cout%256==0) ==')
for Cin, Cout, H in SHAPES:
    if Cout % 256 != 0:
        continue
    x = torch.randn(B, Cin, H, H).to(DEV, torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Cout, Cin, 3, 3) * 0.02).to(DEV, torch.bfloat16)
    bias = torch.randn(Cout).to(DEV, torch.bfloat16)
    w_ohwi = w.permute(0, 2, 3, 1).contiguous().reshape(Cout, 9 * Cin)
    xp = _C.pad1_nhwc(x)
    y = _C.conv_igemm_8p_fwd(xp, w_ohwi, bias, H, H, True)
    ref = torch.relu(torch.nn.functional.conv2d(
        x.float(), w.float(), bias.float(), padding=1))
    rel = ((y.float() - ref).abs().max() /
           ref.abs().max().clamp_min(1e-6)).item()
    t_pad = timeit(lambda: _C.pad1_nhwc(x))
    t_8 = timeit(lambda: _C.conv_igemm_8p_fwd(xp, w_ohwi, bias, H, H,
                                              True))
    gf = 2.0 * B * H * H * Cout * Cin * 9 / 1e9
    print('Cin%4d Cout%4d H%4d  rel %.4f  8p %7.1fus (%5.0f TF) '
          '+pad %5.1fus' % (Cin, Cout, H, rel, t_8, gf / t_8 * 1e6, t_pad))

print('== glds64 variant (padded, Cout=64 class) ==')
for Cin, Cout, H in SHAPES:
    if Cout > 128:
        continue
    x = torch.randn(B, Cin, H, H).to(DEV, torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    w = (torch.randn(Cout, Cin, 3, 3) * 0.02).to(DEV, torch.bfloat16)
    bias = torch.randn(Cout).to(DEV, torch.bfloat16)
    w_ohwi = w.permute(0, 2, 3, 1).contiguous().reshape(Cout, 9 * Cin)
    xp = _C.pad1_nhwc(x)
    y = _C.conv_igemm_glds64_fwd(xp, w_ohwi, bias, H, H, True)
    ref = torch.relu(torch.nn.functional.conv2d(
        x.float(), w.float(), bias.float(), padding=1))
    rel = ((y.float() - ref).abs().max() /
           ref.abs().max().clamp_min(1e-6)).item()
    t_pad = timeit(lambda: _C.pad1_nhwc(x))
    t_g = timeit(lambda: _C.conv_igemm_glds64_fwd(xp, w_ohwi, bias, H, H,
                                                  True))
    gf = 2.0 * B * H * H * Cout * Cin * 9 / 1e9
    print('Cin%4d Cout%4d H%4d  rel %.4f  glds64 %7.1fus (%5.0f TF) '
          '+pad %5.1fus' % (Cin, Cout, H, rel, t_g, gf / t_g * 1e6, t_pad))
