// Hand-written 3x3/s1/pad1 NHWC conv WEIGHT-gradient (wgrad) kernel for
// --train_cnn (reference model.py:505-511 implicit backward surface).
//
// dW[co][ky][kx][ci] = sum_m dy[m][co] * xpad[base(m) + (ky-1,kx-1)][ci]
// viewed as ONE GEMM  C[Cout][9*Cin] = dy^T [Cout][M] @ Xcol [M][9*Cin]
// whose output IS the OHWI weight layout the forward kernels consume.
// Both operands are K(=m)-major in memory, so each 64-m chunk is staged
// TRANSPOSED into LDS ([c][m] image, +8 element row padding to spread
// the scalar-write banks) and the MFMA fragments then read contiguous
// k runs.  Split-K over m-chunks with an fp32 atomicAdd epilogue fills
// the chip on small-tile layers (conv1/2) where (9Cin/64)x(Cout/64)
// blocks alone would not.
//
// dgrad needs no new kernel: it is the same 3x3/s1 conv as forward with
// the weight flipped and io-transposed (sat_amd/ops/convgrad.py routes
// it through conv_igemm_8p / glds / igemm64).

#include "common.h"

namespace {
constexpr int PAD = 8;                 // LDS row pad (bf16 elements)
constexpr int LROW = 64 + PAD;         // LDS row stride
}

__global__ __launch_bounds__(256)
void conv3x3_wgrad_kernel(const bf16* __restrict__ xpad, // [B,H+2,W+2,Ci]
                          const bf16* __restrict__ dy,   // [M, Cout] NHWC
                          float* __restrict__ dw,        // [Cout, 9*Cin]
                          int M, int Hh, int Ww, int Cin, int Cout,
                          int chunks_per, int nchunks) {
    __shared__ bf16 lds[2 * 64 * LROW];
    bf16* A = lds;                     // [64 co][LROW m]
    bf16* Bx = lds + 64 * LROW;        // [64 n ][LROW m]

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 1, wc = wave & 1;
    const int n0 = blockIdx.x * 64;    // within 9*Cin
    const int co0 = blockIdx.y * 64;
    const int dxy = n0 / Cin;          // this block's (ky,kx) plane
    const int ci0 = n0 % Cin;          // 64-aligned since Cin % 64 == 0
    const int dyy = dxy / 3 - 1, dxx = dxy % 3 - 1;
    const int Wp = Ww + 2;
    const int HW = Hh * Ww;

    floatx4 acc[2][2];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

    // this thread's staging slice: rows mrow, mrow+32 of the 64-m chunk,
    // 16-B chunk c8 of the 64-channel tile
    const int mrow = tid >> 3;         // 0..31
    const int c8 = (tid & 7) * 8;

    const int c_begin = blockIdx.z * chunks_per;
    const int c_end = min(c_begin + chunks_per, nchunks);

    for (int ch = c_begin; ch < c_end; ++ch) {
        const int m0 = ch * 64;
#pragma unroll
        for (int half = 0; half < 2; ++half) {
            const int mi = mrow + half * 32;
            const int m = m0 + mi;
            bf16x8 vdy = {};
            bf16x8 vx = {};
            if (m < M) {
                vdy = *(const bf16x8*)(dy + (int64_t)m * Cout + co0 + c8);
                const int b = m / HW;
                const int yx = m % HW;
                const int y = yx / Ww, x = yx % Ww;
                const int64_t base =
                    (((int64_t)b * (Hh + 2) + y + 1 + dyy) * Wp
                     + x + 1 + dxx) * Cin;
                vx = *(const bf16x8*)(xpad + base + ci0 + c8);
            }
            // transposed scalar writes: LDS[c][m]
#pragma unroll
            for (int e = 0; e < 8; ++e) {
                A[(c8 + e) * LROW + mi] = vdy[e];
                Bx[(c8 + e) * LROW + mi] = vx[e];
            }
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            const int kof = kk * 32 + kgrp * 8;
            bf16x8 af[2], bf[2];
#pragma unroll
            for (int i = 0; i < 2; ++i) {
                af[i] = *(const bf16x8*)(
                    A + (wr * 32 + i * 16 + lrow) * LROW + kof);
                bf[i] = *(const bf16x8*)(
                    Bx + (wc * 32 + i * 16 + lrow) * LROW + kof);
            }
#pragma unroll
            for (int i = 0; i < 2; ++i)
#pragma unroll
                for (int j = 0; j < 2; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[i], bf[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }

    // fp32 atomic epilogue (dw zero-initialized by the host wrapper)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        const int col = n0 + wc * 32 + j * 16 + (lane & 15);
#pragma unroll
        for (int i = 0; i < 2; ++i) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = co0 + wr * 32 + i * 16 + (lane >> 4) * 4
                    + r;
                if (row < Cout)
                    atomicAdd(dw + (int64_t)row * 9 * Cin + col,
                              acc[i][j][r]);
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

    auto dw = at::zeros({Cout, 9 * Cin},
                        xpad.options().dtype(at::kFloat));
    int nchunks = cdiv(M, 64);
    int tiles = (9 * Cin / 64) * (Cout / 64);
    // fill ~2 blocks/CU (the kernel is 2-wave-tile, high occupancy)
    int splitk = std::max(1, std::min(32, 512 / tiles));
    int chunks_per = cdiv(nchunks, splitk);
    splitk = cdiv(nchunks, chunks_per);
    dim3 grid(9 * Cin / 64, Cout / 64, splitk);
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(conv3x3_wgrad_kernel, grid, dim3(256), 0, s,
                       (const bf16*)xpad.data_ptr(),
                       (const bf16*)dy_rows.data_ptr(),
                       (float*)dw.data_ptr(), M, (int)Hh, (int)Ww,
                       (int)Cin, (int)Cout, chunks_per, nchunks);
    HIP_OK(hipGetLastError());
    return dw;
}
