// Hand-written 3x3/s1/pad1 NHWC conv WEIGHT-gradient (wgrad) kernel for
// --train_cnn (reference model.py:505-511 implicit backward surface).
//
// dW[co][ky][kx][ci] = sum_m dy[m][co] * xpad[base(m) + (ky-1,kx-1)][ci]
// viewed as ONE GEMM  C[Cout][9*Cin] = dy^T [Cout][M] @ Xcol [M][9*Cin]
// whose output IS the OHWI weight layout the forward kernels consume.
// Both operands are K(=m)-major in memory, so each 64-m chunk is staged
// TRANSPOSED into LDS: each thread loads 8 consecutive-m rows of one
// 8-channel slice and repacks them in registers so every LDS store is a
// full ds_write_b128 (8 m-values of one channel) — no scalar-write
// transpose tax.  Waves 0-1 stage dy, waves 2-3 stage x (each k-chunk =
// 8 loads + 8 b128 writes per thread).  128x128 output tile, 4 waves,
// +8-element LDS row padding for conflict-free transposed reads.
// Split-K over m-chunks with an fp32 atomicAdd epilogue (plain stores
// when the grid needs no split) fills the chip on small-tile layers.
//
// dgrad needs no new kernel: it is the same 3x3/s1 conv as forward with
// the weight flipped and io-transposed (sat_amd/ops/convgrad.py routes
// it through conv_igemm_8p / glds / igemm64).

#include "common.h"

namespace {
constexpr int PAD = 8;                 // LDS row pad (bf16 elements)
constexpr int LROW = 64 + PAD;         // LDS row stride (m-extent 64)
}

// TCO = co-tile rows (128, or 64 for Cout=64 layers where a 128-row
// tile wastes half its loads and MFMA on clamped duplicates)
template <int TCO>
__global__ __launch_bounds__(256)
void conv3x3_wgrad_kernel(const bf16* __restrict__ xpad, // [B,H+2,W+2,Ci]
                          const bf16* __restrict__ dy,   // [M, Cout] NHWC
                          float* __restrict__ dw,        // [Cout, 9*Cin]
                          int M, int Hh, int Ww, int Cin, int Cout,
                          int chunks_per, int nchunks, int use_atomic) {
    __shared__ bf16 lds[(128 + TCO) * LROW];
    bf16* A = lds;                      // [TCO co][LROW m]
    bf16* Bx = lds + TCO * LROW;        // [128 n ][LROW m]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    // wave grid: TCO=128 -> 2(co) x 2(n); TCO=64 -> 1 x 4
    const int wr = (TCO == 128) ? (wave >> 1) : 0;
    const int wc = (TCO == 128) ? (wave & 1) : (wave >> 1);
    const int wn32 = (TCO == 128) ? 0 : (wave & 1);  // 32-col sub-split
    const int n0 = blockIdx.x * 128;    // within 9*Cin
    const int co0 = blockIdx.y * TCO;
    const int Wp = Ww + 2;
    const int HW = Hh * Ww;
    const int N9 = 9 * Cin;

    // ---- staging role of this thread (constant): the first TCO
    // threads stage dy (8 m-rows x 8 channels each), the next 128
    // stage x ----
    const bool stage_dy = (TCO == 128) ? (wave < 2) : (tid < 64);
    const int st = (TCO == 128) ? (tid & 127)
                                : (stage_dy ? tid : (tid - 64) & 127);
    const int snch = stage_dy ? (TCO / 8) : 16;   // 8-ch slices per row
    const int sc8 = (st % snch) * 8;    // channel slice [sc8, sc8+8)
    const int sm8 = (st / snch) * 8;    // m slice [sm8, sm8+8) of chunk
    // clamped global channel base (epilogue guards discard duplicates)
    int dyco = co0 + sc8;
    if (dyco > Cout - 8) dyco = Cout - 8;
    int nidx = n0 + sc8;
    if (nidx > N9 - 8) nidx = N9 - 8;
    const int sdxy = nidx / Cin;        // 8-chunk never straddles a dxy
    const int sci = nidx % Cin;
    const int sdy = sdxy / 3 - 1, sdx = sdxy % 3 - 1;

    floatx4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

    const int c_begin = blockIdx.z * chunks_per;
    const int c_end = min(c_begin + chunks_per, nchunks);

    for (int ch = c_begin; ch < c_end; ++ch) {
        const int m0 = ch * 64 + sm8;   // this thread's first m
        const bool full = (m0 + 7 < M);  // hoisted guard: per-element
        bf16x8 v[8];                     // load-selects de-pipeline
        const bf16x8 vz = {};            // (guide §5 trap c)
        if (stage_dy) {
            const bf16* p = dy + (int64_t)m0 * Cout + dyco;
            if (full) {
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    v[e] = *(const bf16x8*)p;
                    p += Cout;
                }
            } else {
                for (int e = 0; e < 8; ++e) {
                    v[e] = (m0 + e < M) ? *(const bf16x8*)p : vz;
                    p += Cout;
                }
            }
        } else {
            // incremental padded-image base over 8 consecutive pixels
            int b = m0 / HW, yx = m0 % HW;
            int y = yx / Ww, x = yx % Ww;
            const bf16* p = xpad
                + (((int64_t)b * (Hh + 2) + y + 1 + sdy) * Wp
                   + x + 1 + sdx) * Cin + sci;
            if (full) {
#pragma unroll
                for (int e = 0; e < 8; ++e) {
                    v[e] = *(const bf16x8*)p;
                    ++x;
                    if (x == Ww) {      // wrap to next image row
                        x = 0; ++y;
                        if (y == Hh) {  // wrap to next batch image
                            y = 0;
                            p += (int64_t)3 * Wp * Cin
                                - (Ww - 1) * Cin;
                        } else {
                            p += (int64_t)3 * Cin;  // 2 pad cols + 1
                        }
                    } else {
                        p += Cin;
                    }
                }
            } else {
                for (int e = 0; e < 8; ++e) {
                    v[e] = (m0 + e < M) ? *(const bf16x8*)p : vz;
                    ++x;
                    if (x == Ww) {
                        x = 0; ++y;
                        if (y == Hh) {
                            y = 0;
                            p += (int64_t)3 * Wp * Cin
                                - (Ww - 1) * Cin;
                        } else {
                            p += (int64_t)3 * Cin;
                        }
                    } else {
                        p += Cin;
                    }
                }
            }
        }
        __syncthreads();                // previous MFMA done before write
        if (TCO == 128 || tid < 64 + 128) {
            bf16* dst = (stage_dy ? A : Bx) + (int64_t)sc8 * LROW + sm8;
            // repack: row e of LDS gets channel e across the 8 m's
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                bf16x8 o;
#pragma unroll
                for (int mth = 0; mth < 8; ++mth)
                    o[mth] = v[mth][e];
                *(bf16x8*)(dst + e * LROW) = o;
            }
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            const int kof = kk * 32 + kgrp * 8;
            // per-wave output: TCO=128 -> 64co x 64n (4x4 frags);
            // TCO=64 -> 64co x 32n (4x2 frags, n split by wn32)
            constexpr int NJ = (TCO == 128) ? 4 : 2;
            bf16x8 af[4], bf[NJ];
#pragma unroll
            for (int i = 0; i < 4; ++i)
                af[i] = *(const bf16x8*)(
                    A + (wr * 64 + i * 16 + lrow) * LROW + kof);
#pragma unroll
            for (int j = 0; j < NJ; ++j)
                bf[j] = *(const bf16x8*)(
                    Bx + (wc * 64 + wn32 * 32 + j * 16 + lrow) * LROW
                    + kof);
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < NJ; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
    }

    // epilogue: atomics only when K was split (dw pre-zeroed by host)
#pragma unroll
    for (int j = 0; j < ((TCO == 128) ? 4 : 2); ++j) {
        const int col = n0 + wc * 64 + wn32 * 32 + j * 16 + (lane & 15);
        if (col >= N9) continue;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = co0 + wr * 64 + i * 16 + (lane >> 4) * 4
                    + r;
                if (row < Cout) {
                    float* d = dw + (int64_t)row * N9 + col;
                    if (use_atomic) atomicAdd(d, acc[i][j][r]);
                    else *d = acc[i][j][r];
                }
            }
        }
    }
}

at::Tensor conv3x3_wgrad(at::Tensor xpad, at::Tensor dy_rows,
                         int64_t Hh, int64_t Ww) {
    CHECK_GPU(xpad); CHECK_BF16(xpad);
    CHECK_GPU(dy_rows); CHECK_BF16(dy_rows); CHECK_CONTIG(dy_rows);
    TORCH_CHECK(xpad.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = xpad.size(0), Cin = xpad.size(1);
    TORCH_CHECK(xpad.size(2) == Hh + 2 && xpad.size(3) == Ww + 2);
    int Cout = dy_rows.size(1);
    int64_t M64 = (int64_t)B * Hh * Ww;
    TORCH_CHECK(dy_rows.size(0) == M64);
    TORCH_CHECK(Cin % 64 == 0 && Cout % 64 == 0, "wgrad: C % 64");
    TORCH_CHECK(M64 < (1LL << 31));
    int M = (int)M64;

    int nchunks = cdiv(M, 64);
    int tco = (Cout % 128 == 0) ? 128 : 64;
    TORCH_CHECK(Cout % tco == 0, "wgrad: Cout % 64");
    int tiles = cdiv(9 * Cin, 128) * (Cout / tco);
    // target ~2 blocks/CU; the old cap of 32 left small-tile layers at
    // 1 block/CU where the 2-barrier chunk loop is latency-bound
    // (conv1_2 wgrad: 775 us capped vs 386 at full split; big-tile
    // layers prefer 512/tiles — per-shape sweep in r02 logs)
    int splitk = std::max(1, std::min(128, 512 / tiles));
    int chunks_per = cdiv(nchunks, splitk);
    splitk = cdiv(nchunks, chunks_per);
    auto dw = (splitk > 1)
        ? at::zeros({Cout, 9 * Cin}, xpad.options().dtype(at::kFloat))
        : at::empty({Cout, 9 * Cin}, xpad.options().dtype(at::kFloat));
    dim3 grid(cdiv(9 * Cin, 128), Cout / tco, splitk);
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    if (tco == 128)
        hipLaunchKernelGGL((conv3x3_wgrad_kernel<128>), grid, dim3(256),
                           0, s,
                           (const bf16*)xpad.data_ptr(),
                           (const bf16*)dy_rows.data_ptr(),
                           (float*)dw.data_ptr(), M, (int)Hh, (int)Ww,
                           (int)Cin, (int)Cout, chunks_per, nchunks,
                           splitk > 1 ? 1 : 0);
    else
        hipLaunchKernelGGL((conv3x3_wgrad_kernel<64>), grid, dim3(256),
                           0, s,
                           (const bf16*)xpad.data_ptr(),
                           (const bf16*)dy_rows.data_ptr(),
                           (float*)dw.data_ptr(), M, (int)Hh, (int)Ww,
                           (int)Cin, (int)Cout, chunks_per, nchunks,
                           splitk > 1 ? 1 : 0);
    HIP_OK(hipGetLastError());
    return dw;
}
