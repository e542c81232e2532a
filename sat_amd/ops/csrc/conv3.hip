// Direct NHWC 3x3/s1/pad1 convolution for 3-channel input (VGG conv1_1).
//
// MIOpen's NHWC bf16 path falls back to a CK grouped kernel at C_in = 3
// (measured 2.4 ms for [32,224,224,3]->64 — 2.3 TFLOP/s); this layer is
// memory-floor-bound (215 MB traffic ≈ 35 us at HBM rate), so a direct
// kernel wins ~25x.  One thread owns one output pixel and produces all
// C_out channels (contiguous NHWC store, bf16x8); the 3x3x3 input window
// is shared across channels in registers; the [C_out, 27] weights are
// staged once in LDS.
//
// Frozen-CNN forward only (reference trains the CNN only under
// --train_cnn, which uses the autograd conv path).

#include "common.h"

#define MAX_COUT 128
#ifndef LDS_STRIDE
#define LDS_STRIDE 72
#endif

__global__ __launch_bounds__(256)
void conv3_fwd_kernel(const bf16* __restrict__ in,   // [B,H,W,3]
                      const bf16* __restrict__ w,    // [Cout,3,3,3] OIHW
                      const bf16* __restrict__ bias, // [Cout]
                      bf16* __restrict__ out,        // [B,H,W,Cout] or
                                                     // padded variant
                      int BHW, int Hh, int Ww, int Cout, int relu,
                      int emit_pad) {
    __shared__ float ws[27 * MAX_COUT];   // ws[k*Cout + c], k = ci*9+ky*3+kx
    __shared__ float bs[MAX_COUT];
    for (int i = threadIdx.x; i < 27 * Cout; i += blockDim.x) {
        int k = i / Cout, c = i % Cout;
        int ci = k / 9, ky = (k % 9) / 3, kx = k % 3;
        // OIHW: w[c][ci][ky][kx]
        ws[i] = bf2f(w[((c * 3 + ci) * 3 + ky) * 3 + kx]);
    }
    for (int c = threadIdx.x; c < Cout; c += blockDim.x)
        bs[c] = (bias != nullptr) ? bf2f(bias[c]) : 0.f;
    __syncthreads();

    int pix = blockIdx.x * blockDim.x + threadIdx.x;
    if (pix >= BHW) return;
    int b = pix / (Hh * Ww);
    int yx = pix % (Hh * Ww);
    int y = yx / Ww, x = yx % Ww;

    // gather the 3x3x3 input window (zero-padded)
    float win[27];
#pragma unroll
    for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
        for (int kx = 0; kx < 3; ++kx) {
            int yy = y + ky - 1, xx = x + kx - 1;
            bool ok = (yy >= 0 && yy < Hh && xx >= 0 && xx < Ww);
            const bf16* p = in + (((int64_t)b * Hh + yy) * Ww + xx) * 3;
#pragma unroll
            for (int ci = 0; ci < 3; ++ci)
                win[ci * 9 + ky * 3 + kx] = ok ? bf2f(p[ci]) : 0.f;
        }
    }

    // emit_pad: write into the interior of a 1-px zero-bordered buffer
    // so the NEXT 3x3 conv consumes it without a pad pass (the thread
    // already knows (b,y,x) — the padded store is free here, while a
    // separate pad1_nhwc of a 224^2 x64 tensor costs ~70 us)
    bf16* op = emit_pad
        ? out + ((((int64_t)b * (Hh + 2) + y + 1) * (Ww + 2) + x + 1)
                 * Cout)
        : out + (int64_t)pix * Cout;
    for (int c0 = 0; c0 < Cout; c0 += 8) {
        bf16x8 o;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            int c = c0 + e;
            float acc = bs[c];
#pragma unroll
            for (int k = 0; k < 27; ++k)
                acc += win[k] * ws[k * Cout + c];
            if (relu) acc = fmaxf(acc, 0.f);
            o[e] = f2bf(acc);
        }
        *(bf16x8*)(op + c0) = o;
    }
}

// zero only the 1-px border of a padded NHWC buffer (the interior is
// fully overwritten by the producing conv)
__global__ void zero_border_kernel(bf16* __restrict__ out,
                                   int B, int Hp, int Wp, int C) {
    int64_t border = (int64_t)B * 2 * (Hp + Wp - 2);  // border pixels
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t n = border * (C / 8);
    if (i >= n) return;
    int c8 = (int)(i % (C / 8)) * 8;
    int64_t pb = i / (C / 8);
    int b = (int)(pb / (2 * (Hp + Wp - 2)));
    int q = (int)(pb % (2 * (Hp + Wp - 2)));
    int y, x;
    if (q < Wp) { y = 0; x = q; }
    else if (q < 2 * Wp) { y = Hp - 1; x = q - Wp; }
    else if (q < 2 * Wp + (Hp - 2)) { y = q - 2 * Wp + 1; x = 0; }
    else { y = q - 2 * Wp - (Hp - 2) + 1; x = Wp - 1; }
    bf16x8 z = {};
    *(bf16x8*)(out + (((int64_t)b * Hp + y) * Wp + x) * C + c8) = z;
}

at::Tensor conv3_fwd(at::Tensor input, at::Tensor weight, at::Tensor bias,
                     bool relu, bool emit_pad) {
    // input NHWC-contiguous [B,3,H,W] tensor in channels_last
    CHECK_GPU(input); CHECK_BF16(input);
    CHECK_GPU(weight); CHECK_BF16(weight);
    TORCH_CHECK(input.size(1) == 3 && weight.size(1) == 3
                && weight.size(2) == 3 && weight.size(3) == 3,
                "conv3_fwd: expects C_in=3, 3x3 kernel");
    TORCH_CHECK(input.is_contiguous(at::MemoryFormat::ChannelsLast),
                "conv3_fwd: input must be channels_last");
    int B = input.size(0), Hh = input.size(2), Ww = input.size(3);
    int Cout = weight.size(0);
    TORCH_CHECK(Cout % 8 == 0 && Cout <= MAX_COUT);
    auto wc = weight.contiguous();  // OIHW
    auto out = emit_pad
        ? at::empty({B, Cout, Hh + 2, Ww + 2},
                    input.options()
                        .memory_format(at::MemoryFormat::ChannelsLast))
        : at::empty({B, Cout, Hh, Ww},
                    input.options()
                        .memory_format(at::MemoryFormat::ChannelsLast));
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0)
        bias_ptr = (const bf16*)bias.contiguous().data_ptr();
    int BHW = B * Hh * Ww;
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    if (emit_pad) {
        int64_t nb = (int64_t)B * 2 * (Hh + 2 + Ww + 2 - 2) * (Cout / 8);
        hipLaunchKernelGGL(zero_border_kernel, dim3(cdiv(nb, 256)),
                           dim3(256), 0, s,
                           (bf16*)out.data_ptr(), B, Hh + 2, Ww + 2,
                           Cout);
    }
    hipLaunchKernelGGL(conv3_fwd_kernel, dim3(cdiv(BHW, 256)), dim3(256),
                       0, s,
                       (const bf16*)input.data_ptr(),
                       (const bf16*)wc.data_ptr(), bias_ptr,
                       (bf16*)out.data_ptr(), BHW, Hh, Ww, Cout,
                       relu ? 1 : 0, emit_pad ? 1 : 0);
    HIP_OK(hipGetLastError());
    return out;
}

// ---- fused NHWC bias + ReLU (replaces torch's separate bias-add and
// clamp kernels after each MIOpen conv: 2x ~23us -> 1x ~12us per layer) ----

__global__ void bias_act_nhwc_kernel(bf16* __restrict__ y,
                                     const bf16* __restrict__ bias,
                                     int64_t n, int C, int relu) {
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (i8 >= n) return;
    bf16x8 v = *(const bf16x8*)(y + i8);
    int c0 = (int)(i8 % C);
    const bf16x8 bv = *(const bf16x8*)(bias + c0);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        float t = bf2f(v[e]) + bf2f(bv[e]);
        if (relu) t = fmaxf(t, 0.f);
        v[e] = f2bf(t);
    }
    *(bf16x8*)(y + i8) = v;
}

void bias_act_nhwc(at::Tensor y, at::Tensor bias, bool relu) {
    // y: channels_last [B,C,H,W]; per-pixel channel runs are contiguous
    CHECK_GPU(y); CHECK_BF16(y);
    int C = y.size(1);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    TORCH_CHECK(y.is_contiguous(at::MemoryFormat::ChannelsLast));
    int64_t n = y.numel();
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(bias_act_nhwc_kernel, dim3(cdiv(n / 8, 256)),
                       dim3(256), 0, s,
                       (bf16*)y.data_ptr(),
                       (const bf16*)bias.contiguous().data_ptr(),
                       n, C, relu ? 1 : 0);
    HIP_OK(hipGetLastError());
}

// ---- direct NHWC 2x2/s2 max pool (torch's nhwc maxpool measured 74us on
// [32,64,112,112]; traffic floor is ~8us) ----

__global__ void maxpool2x2_nhwc_kernel(const bf16* __restrict__ in,
                                       bf16* __restrict__ out,
                                       int B, int H, int W, int C,
                                       int Ho, int Wo) {
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    int64_t n = (int64_t)B * Ho * Wo * C;
    if (i8 >= n) return;
    int c = (int)(i8 % C);
    int64_t pix = i8 / C;
    int xo = (int)(pix % Wo);
    int64_t t = pix / Wo;
    int yo = (int)(t % Ho);
    int b = (int)(t / Ho);
    int y0 = yo * 2, x0 = xo * 2;
    const bf16* base = in + (((int64_t)b * H + y0) * W + x0) * C + c;
    bf16x8 m = *(const bf16x8*)base;
    bf16x8 v;
    if (x0 + 1 < W) {
        v = *(const bf16x8*)(base + C);
#pragma unroll
        for (int e = 0; e < 8; ++e)
            if (bf2f(v[e]) > bf2f(m[e])) m[e] = v[e];
    }
    if (y0 + 1 < H) {
        v = *(const bf16x8*)(base + (int64_t)W * C);
#pragma unroll
        for (int e = 0; e < 8; ++e)
            if (bf2f(v[e]) > bf2f(m[e])) m[e] = v[e];
        if (x0 + 1 < W) {
            v = *(const bf16x8*)(base + (int64_t)W * C + C);
#pragma unroll
            for (int e = 0; e < 8; ++e)
                if (bf2f(v[e]) > bf2f(m[e])) m[e] = v[e];
        }
    }
    *(bf16x8*)(out + i8) = m;
}

at::Tensor maxpool2x2_nhwc(at::Tensor input) {
    CHECK_GPU(input); CHECK_BF16(input);
    TORCH_CHECK(input.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = input.size(0), C = input.size(1), H = input.size(2),
        W = input.size(3);
    TORCH_CHECK(C % 8 == 0);
    int Ho = (H + 1) / 2, Wo = (W + 1) / 2;
    auto out = at::empty({B, C, Ho, Wo},
                         input.options()
                             .memory_format(at::MemoryFormat::ChannelsLast));
    int64_t n = (int64_t)B * Ho * Wo * C;
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(maxpool2x2_nhwc_kernel, dim3(cdiv(n / 8, 256)),
                       dim3(256), 0, s,
                       (const bf16*)input.data_ptr(),
                       (bf16*)out.data_ptr(), B, H, W, C, Ho, Wo);
    HIP_OK(hipGetLastError());
    return out;
}

// ---- frozen BatchNorm as one NHWC scale/shift (+optional ReLU) kernel:
// y = x*scale + shift with scale = gamma/sqrt(var+eps), shift = beta -
// mean*scale, both precomputed once on the host side ----

__global__ void scale_bias_act_nhwc_kernel(bf16* __restrict__ y,
                                           const bf16* __restrict__ scale,
                                           const bf16* __restrict__ shift,
                                           int64_t n, int C, int relu) {
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (i8 >= n) return;
    bf16x8 v = *(const bf16x8*)(y + i8);
    int c0 = (int)(i8 % C);
    const bf16x8 sv = *(const bf16x8*)(scale + c0);
    const bf16x8 bv = *(const bf16x8*)(shift + c0);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        float t = bf2f(v[e]) * bf2f(sv[e]) + bf2f(bv[e]);
        if (relu) t = fmaxf(t, 0.f);
        v[e] = f2bf(t);
    }
    *(bf16x8*)(y + i8) = v;
}

void scale_bias_act_nhwc(at::Tensor y, at::Tensor scale, at::Tensor shift,
                         bool relu) {
    CHECK_GPU(y); CHECK_BF16(y);
    int C = y.size(1);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    TORCH_CHECK(y.is_contiguous(at::MemoryFormat::ChannelsLast));
    int64_t n = y.numel();
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(scale_bias_act_nhwc_kernel, dim3(cdiv(n / 8, 256)),
                       dim3(256), 0, s,
                       (bf16*)y.data_ptr(),
                       (const bf16*)scale.data_ptr(),
                       (const bf16*)shift.data_ptr(),
                       n, C, relu ? 1 : 0);
    HIP_OK(hipGetLastError());
}

// ---- implicit-GEMM NHWC 3x3/s1/pad1 convolution (Cin % 64 == 0) ----
//
// The conv as a GEMM: M = B*H*W output pixels, N = Cout, K = 9*Cin with
// k ordered (dy, dx, ci), ci fastest.  Weights are repacked once to OHWI
// ([Cout][3][3][Cin]), which IS the [N,K] row-major B operand.  The A
// operand is the input gathered on the fly: each k-tile fixes (dy,dx), so
// every row's address is its cached pixel base plus ONE scalar delta;
// only the edge-validity mask differs per row.  Structure is the tiled
// MFMA GEMM (128x64 tile, 4 waves, LDS +8 padding); bias+ReLU fused in
// the epilogue.

template <int WR, int WC, int FM, int FN>
__global__ __launch_bounds__(256)
void conv_igemm_kernel(const bf16* __restrict__ in,    // [B,H,W,Cin]
                       const bf16* __restrict__ w,     // [Cout,9*Cin] OHWI
                       const bf16* __restrict__ bias,  // [Cout] or null
                       bf16* __restrict__ out,         // [B,H,W,Cout]
                       int M, int Hh, int Ww, int Cin, int Cout,
                       int relu) {
    constexpr int BM = WR * FM * 16;   // 128
    constexpr int BN = WC * FN * 16;   // 64 or 128
    constexpr int BKc = 64;
    __shared__ bf16 As[BM * LDS_STRIDE];
    __shared__ bf16 Bs[BN * LDS_STRIDE];
    __shared__ int rbase[BM];
    __shared__ short rys[BM], rxs[BM];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave / WC;
    const int wc = wave % WC;
    const int bm = blockIdx.y * BM;
    const int bn = blockIdx.x * BN;

    for (int r = tid; r < BM; r += blockDim.x) {
        int m = bm + r;
        if (m < M) {
            int b = m / (Hh * Ww);
            int yx = m % (Hh * Ww);
            int y = yx / Ww, x = yx % Ww;
            rbase[r] = ((b * Hh + y) * Ww + x) * Cin;
            rys[r] = (short)y;
            rxs[r] = (short)x;
        } else {
            rbase[r] = 0;
            rys[r] = -1;
            rxs[r] = -1;
        }
    }
    __syncthreads();

    floatx4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;
    const int KT = 9 * (Cin / BKc);
    constexpr int A_CHUNKS = BM * BKc / 8 / 256;
    constexpr int B_CHUNKS = BN * BKc / 8 / 256;

    for (int kt = 0; kt < KT; ++kt) {
        const int dxy = kt / (Cin / BKc);
        const int ci0 = (kt % (Cin / BKc)) * BKc;
        const int dy = dxy / 3 - 1, dx = dxy % 3 - 1;
        const int delta = (dy * Ww + dx) * Cin + ci0;

#pragma unroll
        for (int i = 0; i < A_CHUNKS; ++i) {
            int q = tid + 256 * i;
            int row = q >> 3;
            int c8 = (q & 7) * 8;
            int yy = rys[row] + dy, xx = rxs[row] + dx;
            bf16x8 v = {};
            if (yy >= 0 && yy < Hh && xx >= 0 && xx < Ww)
                v = *(const bf16x8*)(in + rbase[row] + delta + c8);
            *(bf16x8*)(As + row * LDS_STRIDE + c8) = v;
        }
#pragma unroll
        for (int i = 0; i < B_CHUNKS; ++i) {
            int q = tid + 256 * i;
            int row = q >> 3;
            int c8 = (q & 7) * 8;
            bf16x8 v = {};
            int gb = bn + row;
            if (gb < Cout)
                v = *(const bf16x8*)(
                    w + (int64_t)gb * 9 * Cin + dxy * Cin + ci0 + c8);
            *(bf16x8*)(Bs + row * LDS_STRIDE + c8) = v;
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8 a_frag[FM], b_frag[FN];
            const int kof = kk * 32 + kgrp * 8;
#pragma unroll
            for (int mi = 0; mi < FM; ++mi)
                a_frag[mi] = *(const bf16x8*)(
                    As + (wr * FM * 16 + mi * 16 + lrow) * LDS_STRIDE
                    + kof);
#pragma unroll
            for (int ni = 0; ni < FN; ++ni)
                b_frag[ni] = *(const bf16x8*)(
                    Bs + (wc * FN * 16 + ni * 16 + lrow) * LDS_STRIDE
                    + kof);
#pragma unroll
            for (int mi = 0; mi < FM; ++mi)
#pragma unroll
                for (int ni = 0; ni < FN; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
        }
        __syncthreads();
    }

#pragma unroll
    for (int ni = 0; ni < FN; ++ni) {
        int col = bn + wc * FN * 16 + ni * 16 + (lane & 15);
        float bv = (bias != nullptr && col < Cout) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = bm + wr * FM * 16 + mi * 16 + (lane >> 4) * 4
                    + r;
                if (row < M && col < Cout) {
                    float v = acc[mi][ni][r] + bv;
                    if (relu) v = fmaxf(v, 0.f);
                    out[(int64_t)row * Cout + col] = f2bf(v);
                }
            }
        }
    }
}

at::Tensor conv_igemm_fwd(at::Tensor input, at::Tensor w_ohwi,
                          at::Tensor bias, bool relu) {
    CHECK_GPU(input); CHECK_BF16(input);
    CHECK_GPU(w_ohwi); CHECK_CONTIG(w_ohwi); CHECK_BF16(w_ohwi);
    TORCH_CHECK(input.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = input.size(0), Cin = input.size(1), Hh = input.size(2),
        Ww = input.size(3);
    int Cout = w_ohwi.size(0);
    TORCH_CHECK(Cin % 64 == 0, "conv_igemm: Cin % 64 == 0");
    TORCH_CHECK(w_ohwi.numel() == (int64_t)Cout * 9 * Cin);
    auto out = at::empty({B, Cout, Hh, Ww},
                         input.options()
                             .memory_format(at::MemoryFormat::ChannelsLast));
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0)
        bias_ptr = (const bf16*)bias.contiguous().data_ptr();
    int M = B * Hh * Ww;
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    if (Cout >= 128) {
        dim3 grid(cdiv(Cout, 128), cdiv(M, 128));
        hipLaunchKernelGGL((conv_igemm_kernel<2, 2, 4, 4>), grid,
                           dim3(256), 0, s,
                           (const bf16*)input.data_ptr(),
                           (const bf16*)w_ohwi.data_ptr(), bias_ptr,
                           (bf16*)out.data_ptr(), M, Hh, Ww, Cin, Cout,
                           relu ? 1 : 0);
    } else {
        dim3 grid(cdiv(Cout, 64), cdiv(M, 128));
        hipLaunchKernelGGL((conv_igemm_kernel<4, 1, 2, 4>), grid,
                           dim3(256), 0, s,
                           (const bf16*)input.data_ptr(),
                           (const bf16*)w_ohwi.data_ptr(), bias_ptr,
                           (bf16*)out.data_ptr(), M, Hh, Ww, Cin, Cout,
                           relu ? 1 : 0);
    }
    HIP_OK(hipGetLastError());
    return out;
}

// ---- glds-pipelined implicit-GEMM conv (Cin % 64 == 0, padded input) ----
//
// Same math as conv_igemm_kernel but staged with direct-to-LDS DMA
// (global_load_lds width 16, the gfx950 staging lever) over a 1-pixel
// zero-padded input, so no edge masks exist in the hot loop; each lane
// caches its A/B row base addresses once.  128x128 tile, 4 waves 2x2.

__global__ void pad1_nhwc_kernel(const bf16* __restrict__ in,
                                 bf16* __restrict__ out,
                                 int B, int H, int W, int C) {
    // out: [B, H+2, W+2, C] zero-bordered copy of in
    int Hp = H + 2, Wp = W + 2;
    int64_t n = (int64_t)B * Hp * Wp * C;
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (i8 >= n) return;
    int c = (int)(i8 % C);
    int64_t pix = i8 / C;
    int x = (int)(pix % Wp);
    int64_t t = pix / Wp;
    int y = (int)(t % Hp);
    int b = (int)(t / Hp);
    bf16x8 v = {};
    if (y >= 1 && y <= H && x >= 1 && x <= W)
        v = *(const bf16x8*)(in
                             + (((int64_t)b * H + y - 1) * W + x - 1) * C
                             + c);
    *(bf16x8*)(out + i8) = v;
}

at::Tensor pad1_nhwc(at::Tensor input) {
    CHECK_GPU(input); CHECK_BF16(input);
    TORCH_CHECK(input.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = input.size(0), C = input.size(1), H = input.size(2),
        W = input.size(3);
    TORCH_CHECK(C % 8 == 0);
    auto out = at::empty({B, C, H + 2, W + 2},
                         input.options()
                             .memory_format(at::MemoryFormat::ChannelsLast));
    int64_t n = (int64_t)B * (H + 2) * (W + 2) * C;
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(pad1_nhwc_kernel, dim3(cdiv(n / 8, 256)), dim3(256),
                       0, s,
                       (const bf16*)input.data_ptr(),
                       (bf16*)out.data_ptr(), B, H, W, C);
    HIP_OK(hipGetLastError());
    return out;
}

__global__ __launch_bounds__(256)
void conv_igemm_glds_kernel(const bf16* __restrict__ inp,  // padded NHWC
                            const bf16* __restrict__ w,    // [Cout,9*Cin]
                            const bf16* __restrict__ bias,
                            bf16* __restrict__ out,
                            int M, int Hh, int Ww, int Cin, int Cout,
                            int relu) {
    // output pixel grid is the UNPADDED HxW; input rows live in the
    // padded (H+2)x(W+2) image so every (dy,dx) shift is in bounds.
    constexpr int BM = 128, BN = 128, BKc = 64;
    __shared__ bf16 lds[2 * BM * BKc];    // A then B, lane-linear
    bf16* As = lds;
    bf16* Bs = lds + BM * BKc;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 1, wc = wave & 1;
    const int bm = blockIdx.y * BM;
    const int bn = blockIdx.x * BN;
    const int Wp = Ww + 2;

    // per-lane cached row bases: this lane stages rows r = wave*32 + i*8
    // + (lane>>3) of the A tile and the same rows of the B tile
    int aBase[4];
    int rowA[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int r = wave * 32 + i * 8 + (lane >> 3);
        rowA[i] = r;
        int m = bm + r;
        if (m >= M) m = M - 1;           // clamp: padded reads stay legal
        int b = m / (Hh * Ww);
        int yx = m % (Hh * Ww);
        int y = yx / Ww, x = yx % Ww;
        aBase[i] = ((b * (Hh + 2) + y + 1) * Wp + x + 1) * Cin;
    }
    const bf16* wBase[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int cout = bn + wave * 32 + i * 8 + (lane >> 3);
        if (cout >= Cout) cout = Cout - 1;
        wBase[i] = w + (int64_t)cout * 9 * Cin;
    }
    const int ci8 = (lane & 7) * 8;      // this lane's 16B chunk

    floatx4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;
    const int KT = 9 * (Cin / BKc);

    for (int kt = 0; kt < KT; ++kt) {
        const int dxy = kt / (Cin / BKc);
        const int ci0 = (kt % (Cin / BKc)) * BKc;
        const int dy = dxy / 3 - 1, dx = dxy % 3 - 1;
        const int delta = (dy * Wp + dx) * Cin + ci0 + ci8;

#pragma unroll
        for (int i = 0; i < 4; ++i) {
            // st_16x32 XOR swizzle (guide: bank-spread for ds_read_b128
            // from 128B rows): physical chunk (lane&7) holds LOGICAL
            // chunk (lane&7)^2 on rows with bit2 set, so the per-lane
            // GLOBAL address is pre-swizzled (glds LDS side stays
            // lane-linear) and fragment reads XOR the same bit.
            int swz = (rowA[i] & 4) ? 16 : 0;   // elements (=32 B)
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)
                    (inp + aBase[i] + delta + (ci8 ^ swz) - ci8),
                (__attribute__((address_space(3))) uint32_t*)
                    (As + (wave * 32 + i * 8) * BKc),
                16, 0, 0);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)
                    (wBase[i] + dxy * Cin + ci0 + (ci8 ^ swz)),
                (__attribute__((address_space(3))) uint32_t*)
                    (Bs + (wave * 32 + i * 8) * BKc),
                16, 0, 0);
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8 a_frag[4], b_frag[4];
            const int kof = kk * 32 + kgrp * 8;
#pragma unroll
            for (int mi = 0; mi < 4; ++mi) {
                int r = wr * 64 + mi * 16 + lrow;
                a_frag[mi] = *(const bf16x8*)(
                    As + r * BKc + (kof ^ ((r & 4) ? 16 : 0)));
            }
#pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                int r = wc * 64 + ni * 16 + lrow;
                b_frag[ni] = *(const bf16x8*)(
                    Bs + r * BKc + (kof ^ ((r & 4) ? 16 : 0)));
            }
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
        }
        __syncthreads();
    }

#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
        int col = bn + wc * 64 + ni * 16 + (lane & 15);
        float bv = (bias != nullptr && col < Cout) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = bm + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
                if (row < M && col < Cout) {
                    float vv = acc[mi][ni][r] + bv;
                    if (relu) vv = fmaxf(vv, 0.f);
                    out[(int64_t)row * Cout + col] = f2bf(vv);
                }
            }
        }
    }
}

at::Tensor conv_igemm_glds_fwd(at::Tensor padded, at::Tensor w_ohwi,
                               at::Tensor bias, int64_t Hh, int64_t Ww,
                               bool relu) {
    CHECK_GPU(padded); CHECK_BF16(padded);
    CHECK_GPU(w_ohwi); CHECK_CONTIG(w_ohwi); CHECK_BF16(w_ohwi);
    TORCH_CHECK(padded.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = padded.size(0), Cin = padded.size(1);
    TORCH_CHECK(padded.size(2) == Hh + 2 && padded.size(3) == Ww + 2);
    int Cout = w_ohwi.size(0);
    TORCH_CHECK(Cin % 64 == 0 && Cout % 8 == 0);
    auto out = at::empty({B, Cout, Hh, Ww},
                         padded.options()
                             .memory_format(at::MemoryFormat::ChannelsLast));
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0)
        bias_ptr = (const bf16*)bias.contiguous().data_ptr();
    int M = B * Hh * Ww;
    dim3 grid(cdiv(Cout, 128), cdiv(M, 128));
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(conv_igemm_glds_kernel, grid, dim3(256), 0, s,
                       (const bf16*)padded.data_ptr(),
                       (const bf16*)w_ohwi.data_ptr(), bias_ptr,
                       (bf16*)out.data_ptr(), (int)M, (int)Hh, (int)Ww,
                       (int)Cin, (int)Cout, relu ? 1 : 0);
    HIP_OK(hipGetLastError());
    return out;
}

// ---- BN=64 glds variant for Cout=64 3x3 layers (VGG conv1_2) ----
// Same structure as conv_igemm_glds_kernel but a 128x64 tile with 4
// waves stacked on M (per-wave 32x64 output): the register-staged
// <4,1,2,4> kernel measured 350 TF on conv1_2 while the glds-staged
// 128x128 class reaches 536-658 TF; Cout=64 layers could not use it.

__global__ __launch_bounds__(256)
void conv_igemm_glds64_kernel(const bf16* __restrict__ inp, // padded
                              const bf16* __restrict__ w,   // [Cout,9Ci]
                              const bf16* __restrict__ bias,
                              bf16* __restrict__ out,
                              int M, int Hh, int Ww, int Cin, int Cout,
                              int relu) {
    constexpr int BM = 128, BKc = 64;
    __shared__ bf16 lds[(BM + 64) * BKc];   // A[128][64] + B[64][64]
    bf16* As = lds;
    bf16* Bs = lds + BM * BKc;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;              // 0..3, all stacked on M
    const int bm = blockIdx.y * BM;
    const int bn = blockIdx.x * 64;
    const int Wp = Ww + 2;

    // staging rows: this wave stages A rows wave*32 + i*8 + (lane>>3)
    // (i = 0..3) and B rows wave*16 + j*8 + (lane>>3) (j = 0..1)
    int aBase[4];
    int rowA[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int r = wave * 32 + i * 8 + (lane >> 3);
        rowA[i] = r;
        int m = bm + r;
        if (m >= M) m = M - 1;
        int b = m / (Hh * Ww);
        int yx = m % (Hh * Ww);
        int y = yx / Ww, x = yx % Ww;
        aBase[i] = ((b * (Hh + 2) + y + 1) * Wp + x + 1) * Cin;
    }
    const bf16* wBase[2];
    int rowB[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int r = wave * 16 + j * 8 + (lane >> 3);
        rowB[j] = r;
        int cout = bn + r;
        if (cout >= Cout) cout = Cout - 1;
        wBase[j] = w + (int64_t)cout * 9 * Cin;
    }
    const int ci8 = (lane & 7) * 8;

    floatx4 acc[2][4];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;
    const int KT = 9 * (Cin / BKc);

    for (int kt = 0; kt < KT; ++kt) {
        const int dxy = kt / (Cin / BKc);
        const int ci0 = (kt % (Cin / BKc)) * BKc;
        const int dy = dxy / 3 - 1, dx = dxy % 3 - 1;
        const int delta = (dy * Wp + dx) * Cin + ci0 + ci8;

#pragma unroll
        for (int i = 0; i < 4; ++i) {
            int swz = (rowA[i] & 4) ? 16 : 0;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)
                    (inp + aBase[i] + delta + (ci8 ^ swz) - ci8),
                (__attribute__((address_space(3))) uint32_t*)
                    (As + (wave * 32 + i * 8) * BKc),
                16, 0, 0);
        }
#pragma unroll
        for (int j = 0; j < 2; ++j) {
            int swz = (rowB[j] & 4) ? 16 : 0;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)
                    (wBase[j] + dxy * Cin + ci0 + (ci8 ^ swz)),
                (__attribute__((address_space(3))) uint32_t*)
                    (Bs + (wave * 16 + j * 8) * BKc),
                16, 0, 0);
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
            bf16x8 a_frag[2], b_frag[4];
            const int kof = kk * 32 + kgrp * 8;
#pragma unroll
            for (int mi = 0; mi < 2; ++mi) {
                int r = wave * 32 + mi * 16 + lrow;
                a_frag[mi] = *(const bf16x8*)(
                    As + r * BKc + (kof ^ ((r & 4) ? 16 : 0)));
            }
#pragma unroll
            for (int ni = 0; ni < 4; ++ni) {
                int r = ni * 16 + lrow;
                b_frag[ni] = *(const bf16x8*)(
                    Bs + r * BKc + (kof ^ ((r & 4) ? 16 : 0)));
            }
#pragma unroll
            for (int mi = 0; mi < 2; ++mi)
#pragma unroll
                for (int ni = 0; ni < 4; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
        }
        __syncthreads();
    }

#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
        int col = bn + ni * 16 + (lane & 15);
        float bv = (bias != nullptr && col < Cout) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = bm + wave * 32 + mi * 16 + (lane >> 4) * 4 + r;
                if (row < M && col < Cout) {
                    float vv = acc[mi][ni][r] + bv;
                    if (relu) vv = fmaxf(vv, 0.f);
                    out[(int64_t)row * Cout + col] = f2bf(vv);
                }
            }
        }
    }
}

at::Tensor conv_igemm_glds64_fwd(at::Tensor padded, at::Tensor w_ohwi,
                                 at::Tensor bias, int64_t Hh, int64_t Ww,
                                 bool relu) {
    CHECK_GPU(padded); CHECK_BF16(padded);
    CHECK_GPU(w_ohwi); CHECK_CONTIG(w_ohwi); CHECK_BF16(w_ohwi);
    TORCH_CHECK(padded.is_contiguous(at::MemoryFormat::ChannelsLast));
    int B = padded.size(0), Cin = padded.size(1);
    TORCH_CHECK(padded.size(2) == Hh + 2 && padded.size(3) == Ww + 2);
    int Cout = w_ohwi.size(0);
    TORCH_CHECK(Cin % 64 == 0 && Cout % 8 == 0);
    auto out = at::empty({B, Cout, Hh, Ww},
                         padded.options()
                             .memory_format(at::MemoryFormat::ChannelsLast));
    const bf16* bias_ptr = nullptr;
    if (bias.defined() && bias.numel() > 0)
        bias_ptr = (const bf16*)bias.contiguous().data_ptr();
    int M = B * Hh * Ww;
    dim3 grid(cdiv(Cout, 64), cdiv(M, 128));
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(conv_igemm_glds64_kernel, grid, dim3(256), 0, s,
                       (const bf16*)padded.data_ptr(),
                       (const bf16*)w_ohwi.data_ptr(), bias_ptr,
                       (bf16*)out.data_ptr(), (int)M, (int)Hh, (int)Ww,
                       (int)Cin, (int)Cout, relu ? 1 : 0);
    HIP_OK(hipGetLastError());
    return out;
}
