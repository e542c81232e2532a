#!/usr/bin/env python3
"""Empirical MFMA layout probe: runs dense_fwd on patterned operands and
prints a grid diff vs matmul, so a wrong C/D or A/B lane mapping can be
diagnosed from one GPU run."""

import sys

import torch

sys.path.insert(0, '.')


def main():
    from sat_amd import _C
    dev = 'cuda'
    M = N = K = 32
    # unique-valued rank-1 pattern: Y[i][j] = (i+1)*(100+j)/100
    A = torch.zeros(M, K, device=dev, dtype=torch.bfloat16)
    W = torch.zeros(N, K, device=dev, dtype=torch.bfloat16)
    for i in range(M):
        A[i, 0] = (i + 1) / 16.0
    for j in range(N):
        W[j, 0] = (100 + j) / 100.0
    Y = _C.dense_fwd(A, W, torch.Tensor().to(dev), 0).float()
    ref = (A.float() @ W.float().t())
    err = (Y - ref).abs().max().item()
    print('rank1 max err:', err)
    if err > 1e-2:
        print('Y[0:8,0:8]:\n', Y[:8, :8].cpu().numpy())
        print('ref[0:8,0:8]:\n', ref[:8, :8].cpu().numpy())
        # find where ref[3,5] value landed in Y
        target = ref[3, 5].item()
        loc = (Y - target).abs().argmin().item()
        print('ref[3,5]=%.4f found at Y[%d,%d]' %
              (target, loc // N, loc % N))

    # k-mapping check: A,B nonzero at a single shared k
    A2 = torch.zeros(M, K, device=dev, dtype=torch.bfloat16)
    W2 = torch.zeros(N, K, device=dev, dtype=torch.bfloat16)
    A2[:, 17] = 1.0
    for j in range(N):
        W2[j, 17] = (j + 1) / 32.0
    Y2 = _C.dense_fwd(A2, W2, torch.Tensor().to(dev), 0).float()
    ref2 = A2.float() @ W2.float().t()
    print('k=17 max err:', (Y2 - ref2).abs().max().item())

    # random full check
    torch.manual_seed(0)
    A3 = torch.randn(160, 144, device=dev, dtype=torch.bfloat16)
    W3 = torch.randn(136, 144, device=dev, dtype=torch.bfloat16) * 0.1
    Y3 = _C.dense_fwd(A3, W3, torch.Tensor().to(dev), 0).float()
    ref3 = A3.float() @ W3.float().t()
    rel = ((Y3 - ref3).abs().max() /
           ref3.abs().max().clamp_min(1e-6)).item()
    print('random rel err:', rel)


if __name__ == '__main__':
    main()
