// dense_fwd: Y[M,N] = act(X[M,K] @ W[N,K]^T + bias), bf16 in/out, fp32 acc.
//
// The hand-written MFMA GEMM behind every fc layer of the decoder
// (reference op surface SURVEY.md §2.3: init/attend/decode MLPs, LSTM gate
// GEMM).  CDNA4 structure per the gfx950 playbook:
//   * v_mfma_f32_16x16x32_bf16 tiles, fp32 accumulation in AGPRs;
//   * 128x128 block tile, BK=64, 4 waves of 64x64 each (4x4 fragments);
//   * operands staged through LDS with +8-element row padding so the
//     ds_read_b128 fragment reads spread across banks;
//   * both A and B (weight stored [N,K]) read fragments as contiguous
//     16-byte chunks — no transposes anywhere;
//   * fused epilogue: bias add + tanh/relu + bf16 store.
// Edge tiles (M, N, K not multiples of the tile) are guarded with zero-fill
// loads / masked stores, so any shape with K%8==0 works.

#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define LDS_STRIDE (BK + 8)   // bf16 elements; row stride 144 B (16B-aligned)

#define ACT_NONE 0
#define ACT_TANH 1
#define ACT_RELU 2

__global__ __launch_bounds__(256)
void dense_fwd_kernel(const bf16* __restrict__ A,   // [M,K]
                      const bf16* __restrict__ W,   // [N,K]
                      const bf16* __restrict__ bias, // [N] or nullptr
                      bf16* __restrict__ Y,          // [M,N]
                      int M, int N, int K, int act) {
    __shared__ bf16 As[BM * LDS_STRIDE];
    __shared__ bf16 Bs[BN * LDS_STRIDE];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;           // 0..3
    const int wr = wave >> 1;            // wave row (0..1) -> 64-row slab
    const int wc = wave & 1;             // wave col (0..1) -> 64-col slab

    const int bm = blockIdx.y * BM;
    const int bn = blockIdx.x * BN;

    floatx4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;          // fragment row within 16
    const int kgrp = lane >> 4;          // 0..3 -> 8-element k chunk

    for (int k0 = 0; k0 < K; k0 += BK) {
        // ---- stage A and B tiles (guarded, zero-filled) ----
        // 128 rows x 64 cols = 8192 bf16 / 256 threads = 4 chunks of 8
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            int q = tid + 256 * i;       // chunk id 0..1023
            int row = q >> 3;            // 8 chunks per row
            int c8 = (q & 7) * 8;        // start col within tile
            int gk = k0 + c8;
            bf16x8 av = {};
            int ga = bm + row;
            if (ga < M && gk + 8 <= K) {
                av = *(const bf16x8*)(A + (int64_t)ga * K + gk);
            } else if (ga < M) {
                for (int e = 0; e < 8; ++e)
                    if (gk + e < K) av[e] = A[(int64_t)ga * K + gk + e];
            }
            *(bf16x8*)(As + row * LDS_STRIDE + c8) = av;

            bf16x8 bv = {};
            int gb = bn + row;
            if (gb < N && gk + 8 <= K) {
                bv = *(const bf16x8*)(W + (int64_t)gb * K + gk);
            } else if (gb < N) {
                for (int e = 0; e < 8; ++e)
                    if (gk + e < K) bv[e] = W[(int64_t)gb * K + gk + e];
            }
            *(bf16x8*)(Bs + row * LDS_STRIDE + c8) = bv;
        }
        __syncthreads();

        // ---- MFMA over the two 32-deep k-steps of this tile ----
#pragma unroll
        for (int kk = 0; kk < BK / 32; ++kk) {
            bf16x8 a_frag[4], b_frag[4];
            const int kof = kk * 32 + kgrp * 8;
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
                a_frag[mi] = *(const bf16x8*)(
                    As + (wr * 64 + mi * 16 + lrow) * LDS_STRIDE + kof);
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                b_frag[ni] = *(const bf16x8*)(
                    Bs + (wc * 64 + ni * 16 + lrow) * LDS_STRIDE + kof);
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
        }
        __syncthreads();
    }

    // ---- epilogue: bias + activation + guarded bf16 store ----
    // C/D lane map for 16x16: col = lane&15, row = (lane>>4)*4 + reg.
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
        int col = bn + wc * 64 + ni * 16 + (lane & 15);
        float bv = (bias != nullptr && col < N) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = bm + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
                if (row < M && col < N) {
                    float v = acc[mi][ni][r] + bv;
                    if (act == ACT_TANH) v = tanhf(v);
                    else if (act == ACT_RELU) v = fmaxf(v, 0.f);
                    Y[(int64_t)row * N + col] = f2bf(v);
                }
            }
        }
    }
}

at::Tensor dense_fwd(at::Tensor x, at::Tensor w, at::Tensor bias, int64_t act) {
    CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
    CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
    TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "dense_fwd expects 2-D inputs");
    const int64_t M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(w.size(1) == K, "weight inner dim mismatch");
    TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8 (bf16x8 loads)");
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0) {
        CHECK_GPU(bias); CHECK_CONTIG(bias); CHECK_BF16(bias);
        TORCH_CHECK(bias.numel() == N, "bias size mismatch");
        bias_ptr = (const bf16*)bias.data_ptr();
    }
    auto y = at::empty({M, N}, x.options());
    dim3 grid(cdiv(N, BN), cdiv(M, BM));
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dense_fwd_kernel, grid, dim3(256), 0, stream,
                       (const bf16*)x.data_ptr(), (const bf16*)w.data_ptr(),
                       bias_ptr, (bf16*)y.data_ptr(),
                       (int)M, (int)N, (int)K, (int)act);
    HIP_OK(hipGetLastError());
    return y;
}
