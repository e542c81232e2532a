// 8-phase deep-pipelined implicit-GEMM NHWC 3x3/s1 convolution for the
// Cin>=256 / Cout>=256 VGG layers (conv3_2..conv4_3 shapes) — the layers
// MIOpen's asm igemm previously kept (600-740 TF; our 2-barrier glds
// kernel measured ~470-545 TF, profiles/r01_conv_shapes.txt).
//
// Structure = the CDNA4 guide's 256^2 8-phase GEMM template
// (cdna_hip_programming.md §5 "The 256² 8-phase template") mapped onto
// the conv-as-GEMM view (M = B*H*W pixels, N = Cout, K = 9*Cin with a
// 1-px-padded input so each K-tile's (dy,dx) shift is one scalar address
// delta, no edge masks):
//   * 256x256 output tile, BK=64, 8 waves as 2(M)x4(N), 512 threads;
//   * 128 KiB LDS as a ring of 8 16-KiB half-tile slots; tile t occupies
//     slots (t&1)*4 + {A0,B0,A1,B1} (A half = 128 rows x 64 k, 128-B rows);
//   * staging by global_load_lds dwordx4 (2 per thread per half-tile),
//     one half-tile per phase, issued one full K-tile ahead;
//   * counted s_waitcnt vmcnt(4) per phase — 2 half-tiles stay in flight
//     across raw s_barrier (never __syncthreads: its fence would drain
//     the LDS-DMA queue, the 2-barrier structure's ~20% stall).  The
//     count derivation (vmcnt retires in issue order): the VMW(4)+barrier
//     ending phase k-1 guarantees every wave's stagings S_{<=k-3} have
//     landed, and phase k's quadrant reads touch exactly S_{<=k-3};
//   * per phase one C-quadrant: 16 x mfma_f32_16x16x32_bf16 between
//     s_setprio(1)/(0), fragments via ds_read_b128 with the st_16x32
//     XOR swizzle (source-pre-swizzled global address, lane-linear LDS,
//     read-side XOR — guide rule 21: both-sides-or-neither);
//   * quadrant order (A0B0, A1B0, A1B1, A0B1) matches the staging order
//     so every fragment read lands >= 3 phases after its glds issue.
//
// Per-wave output 128x64 split over both halves of each tile dim:
// rows aH*128 + wr*64 + mi*16, cols bH*128 + wc*32 + ni*16.
//
// Routing (sat_amd/models/nn.py): frozen-CNN forward, Cin % 64 == 0,
// Cout % 256 == 0, padded input from pad1_nhwc.  Row/weight bases are
// int64 (batch-512-per-GPU safe — r01 audit note).

#include "common.h"

namespace {

constexpr int SLOT = 8192;     // bf16 elements per 16-KiB half-tile slot

__device__ __forceinline__ void glds16(const bf16* src, bf16* dst) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)src,
        (__attribute__((address_space(3))) uint32_t*)dst, 16, 0, 0);
}

// st_16x32 swizzle for 128-B LDS rows, in bf16 elements (guide formula
// byte ^= ((byte>>9)&1)<<5 with byte = row*128 + elem*2)
__device__ __forceinline__ int swz16(int row) {
    return (row & 4) ? 16 : 0;
}

}  // namespace

// SPLIT=0: one block owns a tile's whole K; bias/ReLU epilogue to bf16.
// SPLIT=1: grid.z slices the K-tile range; fp32 atomicAdd into scratch
// (zeroed by the host), bias/ReLU applied by f32_bias_act_nhwc — fills
// the chip on low-M shapes (conv5 at batch 32: 50 blocks -> 200).
template <int SPLIT>
__global__ __launch_bounds__(512)
void conv_igemm_8p_kernel(const bf16* __restrict__ inp,  // padded NHWC
                          const bf16* __restrict__ w,    // [Cout,9*Cin]
                          const bf16* __restrict__ bias,
                          bf16* __restrict__ out,
                          float* __restrict__ outf,
                          int M, int Hh, int Ww, int Cin, int Cout,
                          int relu, int kt_per) {
    __shared__ bf16 lds[8 * SLOT];            // ONE shared object (guide
                                              // §5 trap 4a) = 128 KiB

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;                // 0..7
    const int wr = wave >> 2;                 // 0..1 (M)
    const int wc = wave & 3;                  // 0..3 (N)
    const int bm = blockIdx.y * 256;
    const int bn = blockIdx.x * 256;
    const int Wp = Ww + 2;

    // ---- per-thread staging addresses (constant across K-tiles) ----
    // Each phase stages one 128-row half; this thread covers rows
    // j*64 + wave*8 + (lane>>3) (j = 0,1) of the half, 16-B chunk
    // (lane&7) with the source-side XOR pre-swizzle.
    const int srow = wave * 8 + (lane >> 3);  // 0..63
    const int selem = (lane & 7) * 8;         // bf16 chunk base in row
    const bf16* aSrc[2][2];                   // [half][j]
    const bf16* bSrc[2][2];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int r = j * 64 + srow;            // row within half
            int m = bm + h * 128 + r;         // global pixel row
            if (m >= M) m = M - 1;            // clamp (padded reads legal)
            int b = m / (Hh * Ww);
            int yx = m % (Hh * Ww);
            int y = yx / Ww, x = yx % Ww;
            aSrc[h][j] = inp
                + (((int64_t)b * (Hh + 2) + y + 1) * Wp + x + 1) * Cin
                + (selem ^ swz16(r));
            int cb = bn + h * 128 + r;        // global Cout row
            if (cb >= Cout) cb = Cout - 1;
            bSrc[h][j] = w + (int64_t)cb * 9 * Cin + (selem ^ swz16(r));
        }
    }
    // wave-uniform LDS staging bases (glds scatters lane*16B from here)
    bf16* dstJ0 = lds + (wave * 8) * 64;
    bf16* dstJ1 = lds + (64 + wave * 8) * 64;

    const int KT = 9 * (Cin / 64);
    const int kt0 = SPLIT ? blockIdx.z * kt_per : 0;
    const int kt1 = SPLIT ? min(kt0 + kt_per, KT) : KT;

    // per-K-tile source deltas: tile kt has (dxy = kt / (Cin/64),
    // ci0 = (kt % (Cin/64)) * 64)
    const int CT = Cin / 64;

#define A_DELTA(kt) ((((kt) / CT) / 3 - 1) * Wp + ((kt) / CT) % 3 - 1) \
        * Cin + ((kt) % CT) * 64
#define B_DELTA(kt) ((kt) / CT) * Cin + ((kt) % CT) * 64

    // stage half h (0:A0 1:B0 2:A1 3:B1) of K-tile kt
#define STAGE(kt, h) do {                                               \
        int _slot = (((kt) & 1) * 4 + (h));                             \
        bf16* _d0 = dstJ0 + _slot * SLOT;                               \
        bf16* _d1 = dstJ1 + _slot * SLOT;                               \
        if ((h) == 0)      { glds16(aSrc[0][0] + A_DELTA(kt), _d0);     \
                             glds16(aSrc[0][1] + A_DELTA(kt), _d1); }   \
        else if ((h) == 1) { glds16(bSrc[0][0] + B_DELTA(kt), _d0);     \
                             glds16(bSrc[0][1] + B_DELTA(kt), _d1); }   \
        else if ((h) == 2) { glds16(aSrc[1][0] + A_DELTA(kt), _d0);     \
                             glds16(aSrc[1][1] + A_DELTA(kt), _d1); }   \
        else               { glds16(bSrc[1][0] + B_DELTA(kt), _d0);     \
                             glds16(bSrc[1][1] + B_DELTA(kt), _d1); }   \
    } while (0)

    // ---- accumulators: acc[aH*4+mi][bH*2+ni] ----
    floatx4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

    // fragment read helpers: wave's rows wr*64 + mi*16 + lrow of an A
    // half-slot; cols wc*32 + ni*16 + lrow of a B half-slot
#define READ_A(dst, kt, aH) do {                                        \
        const bf16* _s = lds + ((((kt) & 1) * 4 + ((aH) ? 2 : 0)))      \
            * SLOT;                                                     \
        _Pragma("unroll")                                               \
        for (int mi = 0; mi < 4; ++mi) {                                \
            int _r = wr * 64 + mi * 16 + lrow;                          \
            _Pragma("unroll")                                           \
            for (int kk = 0; kk < 2; ++kk)                              \
                dst[mi][kk] = *(const bf16x8*)(                         \
                    _s + _r * 64 + ((kk * 32 + kgrp * 8)                \
                                    ^ swz16(_r)));                      \
        }                                                               \
    } while (0)
#define READ_B(dst, kt, bH) do {                                        \
        const bf16* _s = lds + ((((kt) & 1) * 4 + ((bH) ? 3 : 1)))      \
            * SLOT;                                                     \
        _Pragma("unroll")                                               \
        for (int ni = 0; ni < 2; ++ni) {                                \
            int _r = wc * 32 + ni * 16 + lrow;                          \
            _Pragma("unroll")                                           \
            for (int kk = 0; kk < 2; ++kk)                              \
                dst[ni][kk] = *(const bf16x8*)(                         \
                    _s + _r * 64 + ((kk * 32 + kgrp * 8)                \
                                    ^ swz16(_r)));                      \
        }                                                               \
    } while (0)

#define MFMA16(aH, bH) do {                                             \
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

#define VMW(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
#define BAR() __builtin_amdgcn_s_barrier()
#define LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

    bf16x8 afr[4][2], bfr[2][2];

    // ---- prologue: stage the first K-tile (4 half-tiles, 8 glds per
    // thread), then retire A0+B0 before any wave reads them ----
    STAGE(kt0, 0); STAGE(kt0, 1); STAGE(kt0, 2); STAGE(kt0, 3);
    VMW(4); BAR();

    // ---- steady loop: compute tile t, stage tile t+1 ----
    // Per phase: issue quadrant ds_reads (data's arrival was established
    // by the PREVIOUS phase's VMW+barrier), issue one half-tile glds,
    // counted-wait own stagings, raw barrier (cross-wave publication of
    // the wait), drain lgkm, MFMA the quadrant, barrier (WAR fence for
    // the slot restaged 8 phases after its staging).
    for (int t = kt0; t + 1 < kt1; ++t) {
        // phase 4t: quadrant (A0,B0)
        READ_A(afr, t, 0); READ_B(bfr, t, 0);
        STAGE(t + 1, 0);
        VMW(4); BAR(); LGKM0();
        MFMA16(0, 0);
        BAR();
        // phase 4t+1: quadrant (A1,B0) — reuse bfr
        READ_A(afr, t, 1);
        STAGE(t + 1, 1);
        VMW(4); BAR(); LGKM0();
        MFMA16(1, 0);
        BAR();
        // phase 4t+2: quadrant (A1,B1) — reuse afr
        READ_B(bfr, t, 1);
        STAGE(t + 1, 2);
        VMW(4); BAR(); LGKM0();
        MFMA16(1, 1);
        BAR();
        // phase 4t+3: quadrant (A0,B1) — reuse bfr
        READ_A(afr, t, 0);
        STAGE(t + 1, 3);
        VMW(4); BAR(); LGKM0();
        MFMA16(0, 1);
        BAR();
    }

    // ---- tail tile: everything staged; drain once, then plain phases ----
    VMW(0); BAR();
    {
        const int t = kt1 - 1;
        READ_A(afr, t, 0); READ_B(bfr, t, 0); LGKM0();
        MFMA16(0, 0);
        READ_A(afr, t, 1); LGKM0();
        MFMA16(1, 0);
        READ_B(bfr, t, 1); LGKM0();
        MFMA16(1, 1);
        READ_A(afr, t, 0); LGKM0();
        MFMA16(0, 1);
    }

#undef STAGE
#undef READ_A
#undef READ_B
#undef MFMA16
#undef VMW
#undef BAR
#undef LGKM0
#undef A_DELTA
#undef B_DELTA

    // ---- epilogue: bias + ReLU + guarded store (or split-K atomics) ----
#pragma unroll
    for (int bH = 0; bH < 2; ++bH) {
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
            int col = bn + bH * 128 + wc * 32 + ni * 16 + (lane & 15);
            float bv = (!SPLIT && bias != nullptr && col < Cout)
                ? bf2f(bias[col]) : 0.f;
#pragma unroll
            for (int aH = 0; aH < 2; ++aH) {
#pragma unroll
                for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        int row = bm + aH * 128 + wr * 64 + mi * 16
                            + (lane >> 4) * 4 + r;
                        if (row < M && col < Cout) {
                            if (SPLIT) {
                                atomicAdd(
                                    outf + (int64_t)row * Cout + col,
                                    acc[aH * 4 + mi][bH * 2 + ni][r]);
                            } else {
                                float v = acc[aH * 4 + mi]
                                    [bH * 2 + ni][r] + bv;
                                if (relu) v = fmaxf(v, 0.f);
                                out[(int64_t)row * Cout + col]
                                    = f2bf(v);
                            }
                        }
                    }
                }
            }
        }
    }
}

// fp32 split-K scratch -> bias + act -> bf16 NHWC
__global__ void f32_bias_act_nhwc_kernel(const float* __restrict__ yf,
                                         const bf16* __restrict__ bias,
                                         bf16* __restrict__ out,
                                         int64_t n, int C, int relu) {
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (i8 >= n) return;
    int c0 = (int)(i8 % C);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        float t = yf[i8 + e];
        if (bias != nullptr) t += bf2f(bias[c0 + e]);
        if (relu) t = fmaxf(t, 0.f);
        out[i8 + e] = f2bf(t);
    }
}

at::Tensor conv_igemm_8p_fwd(at::Tensor padded, at::Tensor w_ohwi,
                             at::Tensor bias, int64_t Hh, int64_t Ww,
                             bool relu) {
    CHECK_GPU(padded); CHECK_BF16(padded);
    CHECK_GPU(w_ohwi); CHECK_CONTIG(w_ohwi); CHECK_BF16(w_ohwi);
    TORCH_CHECK(padded.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = padded.size(0), Cin = padded.size(1);
    TORCH_CHECK(padded.size(2) == Hh + 2 && padded.size(3) == Ww + 2);
    int Cout = w_ohwi.size(0);
    TORCH_CHECK(Cin % 64 == 0 && Cin >= 64, "conv_igemm_8p: Cin % 64");
    TORCH_CHECK(Cout % 256 == 0, "conv_igemm_8p: Cout % 256");
    TORCH_CHECK(w_ohwi.numel() == (int64_t)Cout * 9 * Cin);
    auto out = at::empty({B, Cout, Hh, Ww},
                         padded.options()
                             .memory_format(at::MemoryFormat::ChannelsLast));
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0)
        bias_ptr = (const bf16*)bias.contiguous().data_ptr();
    int64_t M64 = (int64_t)B * Hh * Ww;
    TORCH_CHECK(M64 < (1LL << 31), "conv_igemm_8p: M too large");
    int M = (int)M64;
    int blocks = cdiv(Cout, 256) * cdiv(M, 256);
    int KT = 9 * (Cin / 64);
    int splitz = std::max(1, std::min({8, 224 / blocks, KT / 2}));
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    if (splitz > 1) {
        auto scratch = at::zeros({M, Cout},
                                 padded.options().dtype(at::kFloat));
        int kt_per = cdiv(KT, splitz);
        splitz = cdiv(KT, kt_per);
        dim3 grid(cdiv(Cout, 256), cdiv(M, 256), splitz);
        hipLaunchKernelGGL((conv_igemm_8p_kernel<1>), grid, dim3(512),
                           0, s,
                           (const bf16*)padded.data_ptr(),
                           (const bf16*)w_ohwi.data_ptr(), bias_ptr,
                           nullptr, (float*)scratch.data_ptr(),
                           M, (int)Hh, (int)Ww,
                           (int)Cin, (int)Cout, relu ? 1 : 0, kt_per);
        int64_t n = (int64_t)M * Cout;
        hipLaunchKernelGGL(f32_bias_act_nhwc_kernel,
                           dim3(cdiv(n / 8, 256)), dim3(256), 0, s,
                           (const float*)scratch.data_ptr(), bias_ptr,
                           (bf16*)out.data_ptr(), n, (int)Cout,
                           relu ? 1 : 0);
    } else {
        dim3 grid(cdiv(Cout, 256), cdiv(M, 256));
        hipLaunchKernelGGL((conv_igemm_8p_kernel<0>), grid, dim3(512),
                           0, s,
                           (const bf16*)padded.data_ptr(),
                           (const bf16*)w_ohwi.data_ptr(), bias_ptr,
                           (bf16*)out.data_ptr(), nullptr,
                           M, (int)Hh, (int)Ww,
                           (int)Cin, (int)Cout, relu ? 1 : 0, KT);
    }
    HIP_OK(hipGetLastError());
    return out;
}
