// 8-phase deep-pipelined DENSE GEMM (256^2 tile) — the conv8p.hip
// schedule with plain row addressing, for big-M dense shapes where the
// 2-barrier glds tile is prologue-bound (the T1 attention projection
// [T*B*L, 512] x [512, 512] measured ~190 TF there; the identical
// 8-phase structure sustains 630-751 TF on conv).  Same discipline:
// ring of 8 half-tile LDS slots, one glds half-tile per phase staged a
// K-tile ahead, counted vmcnt(4) + raw barriers, setprio around each
// 16-MFMA quadrant, st_16x32 source-side swizzle.  K % 64 == 0,
// N % 8 == 0 (row-clamped), M row-clamped.

#include "common.h"

namespace {

constexpr int SLOT8 = 8192;   // bf16 elements per 16-KiB half-tile slot

__device__ __forceinline__ void glds16d(const bf16* src, bf16* dst) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
}

__device__ __forceinline__ int swz16d(int row) {
    return (row & 4) ? 16 : 0;
}

}  // namespace

__global__ __launch_bounds__(512)
void dense_8p_kernel(const bf16* __restrict__ A,   // [M,K]
                     const bf16* __restrict__ W,   // [N,K]
                     const bf16* __restrict__ bias,
                     bf16* __restrict__ Y,
                     int M, int N, int K, int act) {
    __shared__ bf16 lds[8 * SLOT8];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 2;
    const int wc = wave & 3;
    const int bm = blockIdx.y * 256;
    const int bn = blockIdx.x * 256;

    const int srow = wave * 8 + (lane >> 3);
    const int selem = (lane & 7) * 8;
    const bf16* aSrc[2][2];
    const bf16* bSrc[2][2];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int r = j * 64 + srow;
            int m = bm + h * 128 + r;
            if (m >= M) m = M - 1;
            aSrc[h][j] = A + (int64_t)m * K + (selem ^ swz16d(r));
            int n = bn + h * 128 + r;
            if (n >= N) n = N - 1;
            bSrc[h][j] = W + (int64_t)n * K + (selem ^ swz16d(r));
        }
    }
    bf16* dstJ0 = lds + (wave * 8) * 64;
    bf16* dstJ1 = lds + (64 + wave * 8) * 64;

    const int KT = K / 64;

#define DSTAGE(kt, h) do {                                              \
        int _slot = (((kt) & 1) * 4 + (h));                             \
        bf16* _d0 = dstJ0 + _slot * SLOT8;                              \
        bf16* _d1 = dstJ1 + _slot * SLOT8;                              \
        int _k0 = (kt) * 64;                                            \
        if ((h) == 0)      { glds16d(aSrc[0][0] + _k0, _d0);            \
                             glds16d(aSrc[0][1] + _k0, _d1); }          \
        else if ((h) == 1) { glds16d(bSrc[0][0] + _k0, _d0);            \
                             glds16d(bSrc[0][1] + _k0, _d1); }          \
        else if ((h) == 2) { glds16d(aSrc[1][0] + _k0, _d0);            \
                             glds16d(aSrc[1][1] + _k0, _d1); }          \
        else               { glds16d(bSrc[1][0] + _k0, _d0);            \
                             glds16d(bSrc[1][1] + _k0, _d1); }          \
    } while (0)

    floatx4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

#define DREAD_A(dst, kt, aH) do {                                       \
        const bf16* _s = lds + ((((kt) & 1) * 4 + ((aH) ? 2 : 0)))      \
            * SLOT8;                                                    \
        _Pragma("unroll")                                               \
        for (int mi = 0; mi < 4; ++mi) {                                \
            int _r = wr * 64 + mi * 16 + lrow;                          \
            _Pragma("unroll")                                           \
            for (int kk = 0; kk < 2; ++kk)                              \
                dst[mi][kk] = *(const bf16x8*)(                         \
                    _s + _r * 64 + ((kk * 32 + kgrp * 8)                \
                                    ^ swz16d(_r)));                     \
        }                                                               \
    } while (0)
#define DREAD_B(dst, kt, bH) do {                                       \
        const bf16* _s = lds + ((((kt) & 1) * 4 + ((bH) ? 3 : 1)))      \
            * SLOT8;                                                    \
        _Pragma("unroll")                                               \
        for (int ni = 0; ni < 2; ++ni) {                                \
            int _r = wc * 32 + ni * 16 + lrow;                          \
            _Pragma("unroll")                                           \
            for (int kk = 0; kk < 2; ++kk)                              \
                dst[ni][kk] = *(const bf16x8*)(                         \
                    _s + _r * 64 + ((kk * 32 + kgrp * 8)                \
                                    ^ swz16d(_r)));                     \
        }                                                               \
    } while (0)

#define DMFMA16(aH, bH) do {                                            \
        __builtin_amdgcn_s_setprio(1);                                  \
        _Pragma("unroll")                                               \
        for (int mi = 0; mi < 4; ++mi)                                  \
            _Pragma("unroll")                                           \
            for (int ni = 0; ni < 2; ++ni)                              \
                _Pragma("unroll")                                       \
                for (int kk = 0; kk < 2; ++kk)                          \
                    acc[(aH) * 4 + mi][(bH) * 2 + ni] =                 \
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(        \
                            afr[mi][kk], bfr[ni][kk],                   \
                            acc[(aH) * 4 + mi][(bH) * 2 + ni], 0, 0, 0);\
        __builtin_amdgcn_s_setprio(0);                                  \
    } while (0)

#define DVMW(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
#define DBAR() __builtin_amdgcn_s_barrier()
#define DLGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

    bf16x8 afr[4][2], bfr[2][2];

    DSTAGE(0, 0); DSTAGE(0, 1); DSTAGE(0, 2); DSTAGE(0, 3);
    DVMW(4); DBAR();

    for (int t = 0; t + 1 < KT; ++t) {
        DREAD_A(afr, t, 0); DREAD_B(bfr, t, 0);
        DSTAGE(t + 1, 0);
        DVMW(4); DBAR(); DLGKM0();
        DMFMA16(0, 0);
        DBAR();
        DREAD_A(afr, t, 1);
        DSTAGE(t + 1, 1);
        DVMW(4); DBAR(); DLGKM0();
        DMFMA16(1, 0);
        DBAR();
        DREAD_B(bfr, t, 1);
        DSTAGE(t + 1, 2);
        DVMW(4); DBAR(); DLGKM0();
        DMFMA16(1, 1);
        DBAR();
        DREAD_A(afr, t, 0);
        DSTAGE(t + 1, 3);
        DVMW(4); DBAR(); DLGKM0();
        DMFMA16(0, 1);
        DBAR();
    }

    DVMW(0); DBAR();
    {
        const int t = KT - 1;
        DREAD_A(afr, t, 0); DREAD_B(bfr, t, 0); DLGKM0();
        DMFMA16(0, 0);
        DREAD_A(afr, t, 1); DLGKM0();
        DMFMA16(1, 0);
        DREAD_B(bfr, t, 1); DLGKM0();
        DMFMA16(1, 1);
        DREAD_A(afr, t, 0); DLGKM0();
        DMFMA16(0, 1);
    }

#undef DSTAGE
#undef DREAD_A
#undef DREAD_B
#undef DMFMA16
#undef DVMW
#undef DBAR
#undef DLGKM0

#pragma unroll
    for (int bH = 0; bH < 2; ++bH) {
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
            int col = bn + bH * 128 + wc * 32 + ni * 16 + (lane & 15);
            float bv = (bias != nullptr && col < N)
                ? bf2f(bias[col]) : 0.f;
#pragma unroll
            for (int aH = 0; aH < 2; ++aH) {
#pragma unroll
                for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int row = bm + aH * 128 + wr * 64 + mi * 16
                            + (lane >> 4) * 4 + r;
                        if (row < M && col < N) {
                            float v = acc[aH * 4 + mi][bH * 2 + ni][r]
                                + bv;
                            if (act == 1) v = tanhf(v);
                            else if (act == 2) v = fmaxf(v, 0.f);
                            Y[(int64_t)row * N + col] = f2bf(v);
                        }
                    }
                }
            }
        }
    }
}

at::Tensor dense_8p_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                        int64_t act) {
    CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
    CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
    int64_t M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(K % 64 == 0 && N % 8 == 0 && K >= 128);
    auto y = at::empty({M, N}, x.options());
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0)
        bias_ptr = (const bf16*)bias.data_ptr();
    dim3 grid(cdiv(N, 256), cdiv(M, 256));
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dense_8p_kernel, grid, dim3(512), 0, s,
                       (const bf16*)x.data_ptr(),
                       (const bf16*)w.data_ptr(), bias_ptr,
                       (bf16*)y.data_ptr(), (int)M, (int)N, (int)K,
                       (int)act);
    HIP_OK(hipGetLastError());
    return y;
}
