import sys
import torch
sys.path.insert(0, '.')
from sat_amd import _C

DEV = 'cuda'
B, L, A = 32, 196, 512
tdrop = torch.randn(B * L, A, device=DEV, dtype=torch.bfloat16)
v = torch.randn(A, device=DEV, dtype=torch.bfloat16)
dlogits = torch.randn(B, L, device=DEV)
seed = torch.tensor(1, dtype=torch.int64, device=DEV)
dv = torch.zeros(A, device=DEV)


def time_it(fn, iters=200):
    for _ in range(20):
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


us = time_it(lambda: _C.attn_scores_bwd_acc(tdrop, v, dlogits, seed,
                                            0.5, 3, L, dv))
print('scores_bwd p=0.5: %.1f us' % us)
us = time_it(lambda: _C.attn_scores_bwd_acc(tdrop, v, dlogits, seed,
                                            0.0, 3, L, dv))
print('scores_bwd p=0.0: %.1f us' % us)

# compare with pure-traffic ops of same footprint
x = torch.randn(B * L, A, device=DEV, dtype=torch.bfloat16)
us = time_it(lambda: x.clone())
print('same-size clone (12.8MB traffic): %.1f us' % us)

us = time_it(lambda: _C.attn_scores_fused(x, torch.randn(
    B, A, device=DEV, dtype=torch.bfloat16), v, seed, 0.5, 3, L))
print('scores_fused fwd: %.1f us' % us)

# pool kernels
ctx = torch.randn(B, L, A, device=DEV, dtype=torch.bfloat16)
logits = torch.randn(B, L, device=DEV)
us = time_it(lambda: _C.attn_pool_fwd(ctx, logits))
print('pool_fwd: %.1f us' % us)
al, po = _C.attn_pool_fwd(ctx, logits)
us = time_it(lambda: _C.attn_pool_bwd(ctx, al, logits, po, False))
print('pool_bwd: %.1f us' % us)
