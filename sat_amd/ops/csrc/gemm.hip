// dense_fwd: Y[M,N] = act(X[M,K] @ W[N,K]^T + bias), bf16 in/out, fp32 acc.
//
// The hand-written MFMA GEMM family behind every fc layer of the decoder
// (reference op surface SURVEY.md §2.3: init/attend/decode MLPs, LSTM gate
// GEMM).  Three shapes of work, three kernels:
//
//   * tiled_gemm<2,2,4,4>  — 128x128 tile, the large-M path (attention
//     projection [B·196,512]x[512,512]);
//   * tiled_gemm<4,1,2,4>  — 128x64 tile for N <= 512: twice the blocks,
//     fills the 256-CU / 8-XCD chip at the flagship shapes;
//   * skinny split-K       — M <= 32 (decode/LSTM GEMMs at batch 32):
//     one 32-row tile, 16-column wave slabs, K split across blocks with
//     fp32 atomic accumulation + a fused bias/act epilogue pass.  The
//     128x128 kernel runs these shapes at 8-40 blocks (latency-bound,
//     measured 40-60us each); the skinny kernel runs 128-316 blocks.
//
// CDNA4 structure per the gfx950 playbook: v_mfma_f32_16x16x32_bf16,
// fp32 AGPR accumulation, LDS staging with +8 bf16 row padding (bank
// spread for ds_read_b128 fragment reads), bf16x8 (16 B) global loads,
// guarded edges so any K%8==0 shape works.

#include "common.h"

#include <mutex>
#include <unordered_map>
#include <map>
#include <mutex>

#define BK 64
#define LDS_STRIDE (BK + 8)   // row stride 144 B (16 B aligned)

#define ACT_NONE 0
#define ACT_TANH 1
#define ACT_RELU 2

__device__ __forceinline__ float apply_act(float v, int act) {
    if (act == ACT_TANH) return tanhf(v);
    if (act == ACT_RELU) return fmaxf(v, 0.f);
    return v;
}

// ---------------------------------------------------------------------
// tiled kernel: BM = WR*FM*16, BN = WC*FN*16; 4 waves (WR*WC == 4)
// ---------------------------------------------------------------------

template <int WR, int WC, int FM, int FN>
__global__ __launch_bounds__(256)
void tiled_gemm_kernel(const bf16* __restrict__ A,    // [M,K]
                       const bf16* __restrict__ W,    // [N,K]
                       const bf16* __restrict__ bias, // [N] or null
                       bf16* __restrict__ Y,          // [M,N]
                       int M, int N, int K, int act,
                       // split-K: when Yf != null, block z covers
                       // K-range [kq*kchunk, ...) and stores an fp32
                       // slab; skinny_epilogue_kernel combines
                       float* __restrict__ Yf, int kchunk) {
    constexpr int BM = WR * FM * 16;
    constexpr int BN = WC * FN * 16;
    __shared__ bf16 As[BM * LDS_STRIDE];
    __shared__ bf16 Bs[BN * LDS_STRIDE];

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave / WC;
    const int wc = wave % WC;

    const int bm = blockIdx.y * BM;
    const int bn = blockIdx.x * BN;

    floatx4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

    constexpr int A_CHUNKS = BM * BK / 8 / 256;  // bf16x8 chunks per thread
    constexpr int B_CHUNKS = BN * BK / 8 / 256;

    int kbeg = 0, kend = K;
    if (Yf != nullptr) {
        kbeg = blockIdx.z * kchunk;
        kend = min(K, kbeg + kchunk);
    }
    for (int k0 = kbeg; k0 < kend; k0 += BK) {
#pragma unroll
        for (int i = 0; i < A_CHUNKS; ++i) {
            int q = tid + 256 * i;
            int row = q >> 3;
            int c8 = (q & 7) * 8;
            int gk = k0 + c8;
            bf16x8 v = {};
            int ga = bm + row;
            if (ga < M && gk + 8 <= K) {
                v = *(const bf16x8*)(A + (int64_t)ga * K + gk);
            } else if (ga < M) {
                for (int e = 0; e < 8; ++e)
                    if (gk + e < K) v[e] = A[(int64_t)ga * K + gk + e];
            }
            *(bf16x8*)(As + row * LDS_STRIDE + c8) = v;
        }
#pragma unroll
        for (int i = 0; i < B_CHUNKS; ++i) {
            int q = tid + 256 * i;
            int row = q >> 3;
            int c8 = (q & 7) * 8;
            int gk = k0 + c8;
            bf16x8 v = {};
            int gb = bn + row;
            if (gb < N && gk + 8 <= K) {
                v = *(const bf16x8*)(W + (int64_t)gb * K + gk);
            } else if (gb < N) {
                for (int e = 0; e < 8; ++e)
                    if (gk + e < K) v[e] = W[(int64_t)gb * K + gk + e];
            }
            *(bf16x8*)(Bs + row * LDS_STRIDE + c8) = v;
        }
        __syncthreads();

#pragma unroll
        for (int kk = 0; kk < BK / 32; ++kk) {
            bf16x8 a_frag[FM], b_frag[FN];
            const int kof = kk * 32 + kgrp * 8;
#pragma unroll
            for (int mi = 0; mi < FM; ++mi)
                a_frag[mi] = *(const bf16x8*)(
                    As + (wr * FM * 16 + mi * 16 + lrow) * LDS_STRIDE + kof);
#pragma unroll
            for (int ni = 0; ni < FN; ++ni)
                b_frag[ni] = *(const bf16x8*)(
                    Bs + (wc * FN * 16 + ni * 16 + lrow) * LDS_STRIDE + kof);
#pragma unroll
            for (int mi = 0; mi < FM; ++mi)
#pragma unroll
                for (int ni = 0; ni < FN; ++ni)
                    acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
        }
        __syncthreads();
    }

    // epilogue (C/D map: col = lane&15, row = (lane>>4)*4 + reg)
    float* slab = (Yf != nullptr)
        ? Yf + (int64_t)blockIdx.z * M * N : nullptr;
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) {
        int col = bn + wc * FN * 16 + ni * 16 + (lane & 15);
        float bv = (bias != nullptr && col < N) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < FM; ++mi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = bm + wr * FM * 16 + mi * 16 + (lane >> 4) * 4 + r;
                if (row < M && col < N) {
                    if (slab != nullptr)
                        slab[(int64_t)row * N + col] = acc[mi][ni][r];
                    else
                        Y[(int64_t)row * N + col] =
                            f2bf(apply_act(acc[mi][ni][r] + bv, act));
                }
            }
        }
    }
}

// ---------------------------------------------------------------------
// skinny kernel: M <= 32; grid (ceil(N/64), SPLITK); 4 waves of 16 cols;
// A fragments straight from global (x is L2-resident and tiny), W
// fragments streamed from global.  SPLITK == 1 stores the final bf16;
// otherwise fp32 atomics into a workspace + separate epilogue.
// ---------------------------------------------------------------------

template <int RF>
__global__ __launch_bounds__(256)
void skinny_gemm_kernel(const bf16* __restrict__ A,   // [M,K], M <= RF*16
                        const bf16* __restrict__ W,   // [N,K]
                        const bf16* __restrict__ bias,
                        bf16* __restrict__ Y,         // [M,N]
                        float* __restrict__ Yf,       // slabs (splitk>1)
                        uint32_t* __restrict__ cnt,   // per-tile monotonic
                        int M, int N, int K, int act, int splitk) {
    __shared__ uint32_t last_flag;    // the ONLY shared object
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int n0 = blockIdx.x * 64 + wave * 16;
    const bool active = (n0 < N);

    const int kq = blockIdx.y;
    const int kchunk = ((K / 32 + splitk - 1) / splitk) * 32;
    const int kbeg = kq * kchunk;
    const int kend = min(K, kbeg + kchunk);

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

    floatx4 acc[RF];
#pragma unroll
    for (int i = 0; i < RF; ++i)
        acc[i] = floatx4{0.f, 0.f, 0.f, 0.f};

    const bool wfull = (n0 + 16 <= N);
    if (active) {
        for (int k = kbeg; k < kend; k += 32) {
            const int kof = k + kgrp * 8;
            bf16x8 b_frag = {};
            if (kof + 8 <= K) {
                int gb = n0 + lrow;
                if (wfull || gb < N)
                    b_frag = *(const bf16x8*)(W + (int64_t)gb * K + kof);
            } else if (kof < K) {
                int gb = n0 + lrow;
                if (wfull || gb < N)
                    for (int e = 0; e < 8 && kof + e < K; ++e)
                        b_frag[e] = W[(int64_t)gb * K + kof + e];
            }
#pragma unroll
            for (int mi = 0; mi < RF; ++mi) {
                bf16x8 a0 = {};
                int ar = mi * 16 + lrow;
                if (kof + 8 <= K) {
                    if (ar < M)
                        a0 = *(const bf16x8*)(A + (int64_t)ar * K + kof);
                } else if (kof < K) {
                    if (ar < M)
                        for (int e = 0; e < 8 && kof + e < K; ++e)
                            a0[e] = A[(int64_t)ar * K + kof + e];
                }
                acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0, b_frag, acc[mi], 0, 0, 0);
            }
        }
    }

    const int col = n0 + (lane & 15);
    if (splitk == 1) {
        if (!active || col >= N) return;
        float bv = (bias != nullptr) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < RF; ++mi)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = mi * 16 + (lane >> 4) * 4 + r;
                if (row < M)
                    Y[(int64_t)row * N + col] =
                        f2bf(apply_act(acc[mi][r] + bv, act));
            }
        return;
    }

    // ---- split-K slab store + in-launch combine (agent-scope
    // release/acquire hand-off, monotonic per-tile ticket counter) ----
    if (active && col < N) {
        float* slab = Yf + (int64_t)kq * M * N;
#pragma unroll
        for (int mi = 0; mi < RF; ++mi)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = mi * 16 + (lane >> 4) * 4 + r;
                if (row < M)
                    slab[(int64_t)row * N + col] = acc[mi][r];
            }
    }
    if (cnt == nullptr) return;   // slab mode: a separate epilogue reduces
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    if (threadIdx.x == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        uint32_t old = __hip_atomic_fetch_add(
            &cnt[blockIdx.x], 1u, __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_AGENT);
        last_flag = (((old + 1) % (uint32_t)splitk) == 0u) ? 1u : 0u;
    }
    __syncthreads();
    if (last_flag == 0u) return;
    if (threadIdx.x == 0)
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    __syncthreads();

    // reducer: this block sums all slabs for its 64-column tile
    int ncols = min(64, N - blockIdx.x * 64);
    int nel = M * ncols;
    for (int i = threadIdx.x; i < nel; i += (int)blockDim.x) {
        int row = i / ncols;
        int c = blockIdx.x * 64 + (i % ncols);
        float vsum = 0.f;
        for (int q = 0; q < splitk; ++q)
            vsum += Yf[(int64_t)q * M * N + (int64_t)row * N + c];
        if (bias != nullptr) vsum += bf2f(bias[c]);
        Y[(int64_t)row * N + c] = f2bf(apply_act(vsum, act));
    }
}

__global__ void skinny_epilogue_kernel(const float* __restrict__ Yf,
                                       const bf16* __restrict__ bias,
                                       bf16* __restrict__ Y,
                                       int64_t n, int N, int act,
                                       int splitk) {
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    float v = 0.f;
    for (int k = 0; k < splitk; ++k) v += Yf[k * n + idx];
    if (bias != nullptr) v += bf2f(bias[idx % N]);
    Y[idx] = f2bf(apply_act(v, act));
}

__global__ void dense_glds_kernel(const bf16* __restrict__ A,
                                  const bf16* __restrict__ W,
                                  const bf16* __restrict__ bias,
                                  bf16* __restrict__ Y,
                                  int M, int N, int K, int act);

// persistent per-grid ticket counters for the in-launch split-K
// combine (monotonic, modulo-read — never reset, so graph replays and
// repeated launches stay correct; zero-initialized once per size)
static at::Tensor _skinny_cnt(int nblocks, const at::TensorOptions& o) {
    static std::unordered_map<int, at::Tensor> cache;
    static std::mutex mu;
    std::lock_guard<std::mutex> g(mu);
    auto it = cache.find(nblocks);
    if (it == cache.end())
        it = cache.emplace(nblocks,
                           at::zeros({nblocks},
                                     o.dtype(at::kInt))).first;
    return it->second;
}

static bool _skinny_fused() {
    // DEFAULT OFF: the in-launch combine measured 6.87 vs 6.19 ms/step
    // on the flagship (r02 same-box A/B) — at these B=32 shapes the
    // last-arriving block's serial slab read blocks the dependent next
    // kernel, while the separate 256-block epilogue parallelizes it.
    // Kept behind SAT_SKINNY_FUSED=1 for larger-M experiments.
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("SAT_SKINNY_FUSED");
        v = (e != nullptr && e[0] == '1') ? 1 : 0;
    }
    return v == 1;
}

at::Tensor dense_fwd_out(at::Tensor x, at::Tensor w, at::Tensor bias,
                         int64_t act, at::Tensor out) {
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
    at::Tensor y;
    if (out.defined() && out.numel() > 0) {
        CHECK_CONTIG(out); CHECK_BF16(out);
        TORCH_CHECK(out.numel() == M * N, "out size mismatch");
        y = out;
    } else {
        y = at::empty({M, N}, x.options());
    }
    hipStream_t stream = at::cuda::getCurrentCUDAStream();

    if (M <= 128 && K % 32 == 0) {
        // skinny path with split-K chosen to fill the 256-CU chip
        int nblocks = cdiv(N, 64);
        int splitk = 1;
        while (nblocks * splitk < 192 && splitk < 8 &&
               (int)(K / 32) >= 2 * splitk)
            splitk *= 2;
#define LAUNCH_SKINNY(RF, GRID, BIAS, YPTR, YFPTR, SPLITK) \
        hipLaunchKernelGGL((skinny_gemm_kernel<RF>), GRID, dim3(256), 0, \
                           stream, (const bf16*)x.data_ptr(), \
                           (const bf16*)w.data_ptr(), BIAS, YPTR, YFPTR, \
                           nullptr, (int)M, (int)N, (int)K, (int)act, \
                           SPLITK)
        if (splitk == 1) {
            if (M <= 32)
                LAUNCH_SKINNY(2, dim3(nblocks, 1), bias_ptr,
                              (bf16*)y.data_ptr(), nullptr, 1);
            else if (M <= 64)
                LAUNCH_SKINNY(4, dim3(nblocks, 1), bias_ptr,
                              (bf16*)y.data_ptr(), nullptr, 1);
            else
                LAUNCH_SKINNY(8, dim3(nblocks, 1), bias_ptr,
                              (bf16*)y.data_ptr(), nullptr, 1);
        } else if (_skinny_fused()) {
            // in-launch combine: each 64-col tile's last-arriving block
            // reduces the slabs (agent-scope release/acquire + ticket,
            // §6 G16 recipe) — no epilogue launch
            auto yf = at::empty({splitk, M, N},
                                x.options().dtype(at::kFloat));
            auto cnt = _skinny_cnt(nblocks, x.options());
#define LAUNCH_SKF(RF)             hipLaunchKernelGGL((skinny_gemm_kernel<RF>),                                dim3(nblocks, splitk), dim3(256), 0,                                stream, (const bf16*)x.data_ptr(),                                (const bf16*)w.data_ptr(), bias_ptr,                                (bf16*)y.data_ptr(),                                (float*)yf.data_ptr(),                                (uint32_t*)cnt.data_ptr(), (int)M,                                (int)N, (int)K, (int)act, splitk)
            if (M <= 32) LAUNCH_SKF(2);
            else if (M <= 64) LAUNCH_SKF(4);
            else LAUNCH_SKF(8);
#undef LAUNCH_SKF
        } else {
            auto yf = at::empty({splitk, M, N},
                                x.options().dtype(at::kFloat));
            if (M <= 32)
                LAUNCH_SKINNY(2, dim3(nblocks, splitk), nullptr, nullptr,
                              (float*)yf.data_ptr(), splitk);
            else if (M <= 64)
                LAUNCH_SKINNY(4, dim3(nblocks, splitk), nullptr, nullptr,
                              (float*)yf.data_ptr(), splitk);
            else
                LAUNCH_SKINNY(8, dim3(nblocks, splitk), nullptr, nullptr,
                              (float*)yf.data_ptr(), splitk);
            int64_t n = M * N;
            hipLaunchKernelGGL(skinny_epilogue_kernel,
                               dim3(cdiv(n, 256)), dim3(256), 0, stream,
                               (const float*)yf.data_ptr(), bias_ptr,
                               (bf16*)y.data_ptr(), n, (int)N, (int)act,
                               splitk);
        }
#undef LAUNCH_SKINNY
    } else if (M >= 16384 && N >= 256 && N % 8 == 0 && K % 64 == 0
               && K >= 128 && !(out.defined() && out.numel() > 0)) {
        // big-M dense shapes: the 8-phase deep-pipelined 256^2 tile
        // (gemm8p.hip; the 2-barrier glds tile measured only ~190 TF
        // here — short K leaves it prologue-bound)
        extern at::Tensor dense_8p_fwd(at::Tensor, at::Tensor,
                                       at::Tensor, int64_t);
        at::Tensor bb = (bias_ptr != nullptr) ? bias : at::Tensor();
        return dense_8p_fwd(x, w, bb, act);
    } else if (M >= 16384 && N % 128 == 0 && K % 64 == 0) {
        // (retained) glds-staged 128^2 tile
        dim3 grid(cdiv(N, 128), cdiv(M, 128));
        hipLaunchKernelGGL(dense_glds_kernel, grid, dim3(256), 0,
                           stream,
                           (const bf16*)x.data_ptr(),
                           (const bf16*)w.data_ptr(), bias_ptr,
                           (bf16*)y.data_ptr(), (int)M, (int)N, (int)K,
                           (int)act);
    } else {
        // mid-M shapes (the decode-head GEMMs: M=640, N=1024..1536)
        // leave the 2-barrier tile grid at 40-60 blocks on 256 CUs;
        // split K until ~2/3 of the chip has a block (measured 28-42 TF
        // -> the slab+epilogue pair more than doubles it)
        int bn_t = (N <= 512) ? 64 : 128;
        int tiles = cdiv(N, bn_t) * cdiv(M, 128);
        int splitk = 1;
        while (tiles * splitk < 160 && splitk < 8
               && K >= 128 * 2 * splitk)
            splitk *= 2;
        int kchunk = cdiv(cdiv(K, splitk), 64) * 64;
        splitk = cdiv(K, kchunk);
        float* yf_ptr = nullptr;
        at::Tensor yf;
        if (splitk > 1) {
            yf = at::empty({splitk, M, N},
                           x.options().dtype(at::kFloat));
            yf_ptr = (float*)yf.data_ptr();
        }
        dim3 grid(cdiv(N, bn_t), cdiv(M, 128), splitk);
        if (N <= 512)
            hipLaunchKernelGGL((tiled_gemm_kernel<4, 1, 2, 4>), grid,
                               dim3(256), 0, stream,
                               (const bf16*)x.data_ptr(),
                               (const bf16*)w.data_ptr(), bias_ptr,
                               (bf16*)y.data_ptr(), (int)M, (int)N,
                               (int)K, (int)act, yf_ptr, kchunk);
        else
            hipLaunchKernelGGL((tiled_gemm_kernel<2, 2, 4, 4>), grid,
                               dim3(256), 0, stream,
                               (const bf16*)x.data_ptr(),
                               (const bf16*)w.data_ptr(), bias_ptr,
                               (bf16*)y.data_ptr(), (int)M, (int)N,
                               (int)K, (int)act, yf_ptr, kchunk);
        if (splitk > 1) {
            int64_t n = M * N;
            hipLaunchKernelGGL(skinny_epilogue_kernel,
                               dim3(cdiv(n, 256)), dim3(256), 0, stream,
                               (const float*)yf.data_ptr(), bias_ptr,
                               (bf16*)y.data_ptr(), n, (int)N, (int)act,
                               splitk);
        }
    }
    HIP_OK(hipGetLastError());
    return y;
}

at::Tensor dense_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                     int64_t act) {
    return dense_fwd_out(x, w, bias, act, at::Tensor());
}

// ---- GEMM + counter-hash dropout fused into the split-K epilogue ----
// (same hash as hash_dropout_kernel — backward/forward mask pairs must
// regenerate identically)

__device__ __forceinline__ uint32_t mix3g(uint32_t a, uint32_t b,
                                          uint32_t c) {
    uint32_t h = a * 0x9E3779B1u ^ b * 0x85EBCA77u ^ c * 0xC2B2AE3Du;
    h ^= h >> 16; h *= 0x7FEB352Du;
    h ^= h >> 15; h *= 0x846CA68Bu;
    h ^= h >> 16;
    return h;
}

__device__ __forceinline__ float dscaleg(uint32_t seed, int salt,
                                         uint32_t idx, float p) {
    if (p <= 0.f) return 1.f;
    uint32_t h = mix3g(seed, (uint32_t)salt, idx);
    float u = (h >> 8) * (1.0f / 16777216.0f);
    return u >= p ? 1.0f / (1.0f - p) : 0.0f;
}

__global__ void skinny_epilogue_drop_kernel(
        const float* __restrict__ Yf, bf16* __restrict__ Y,
        const int64_t* __restrict__ seed_p, int64_t n, int splitk,
        float p, int salt) {
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    float v = 0.f;
    for (int k = 0; k < splitk; ++k) v += Yf[k * n + idx];
    v *= dscaleg((uint32_t)(*seed_p), salt, (uint32_t)idx, p);
    Y[idx] = f2bf(v);
}

at::Tensor dense_fwd_drop(at::Tensor x, at::Tensor w, at::Tensor seed,
                          double p, int64_t salt) {
    // bias-free, activation-free skinny GEMM whose split-K epilogue
    // applies hash dropout (salt/index convention of hash_dropout) —
    // removes one elementwise launch per decoder reverse step
    CHECK_GPU(x); CHECK_CONTIG(x); CHECK_BF16(x);
    CHECK_GPU(w); CHECK_CONTIG(w); CHECK_BF16(w);
    int64_t M = x.size(0), K = x.size(1), N = w.size(0);
    hipStream_t stream = at::cuda::getCurrentCUDAStream();
    if (p <= 0.0 || M > 128 || K % 32 != 0) {
        // non-skinny shapes: plain GEMM + the separate dropout pass
        // (defined in kernels.hip; same hash)
        extern at::Tensor hash_dropout(at::Tensor, at::Tensor, double,
                                       int64_t);
        auto y = dense_fwd_out(x, w, at::Tensor(), 0, at::Tensor());
        return p > 0.0 ? hash_dropout(y, seed, p, salt) : y;
    }
    auto y = at::empty({M, N}, x.options());
    int nblocks = cdiv(N, 64);
    int splitk = 1;
    while (nblocks * splitk < 192 && splitk < 8 &&
           (int)(K / 32) >= 2 * splitk)
        splitk *= 2;
    auto yf = at::empty({splitk, M, N}, x.options().dtype(at::kFloat));
#define LAUNCH_SK(RF)     hipLaunchKernelGGL((skinny_gemm_kernel<RF>), dim3(nblocks, splitk),                        dim3(256), 0, stream, (const bf16*)x.data_ptr(),                        (const bf16*)w.data_ptr(), nullptr, nullptr,                        (float*)yf.data_ptr(), nullptr, (int)M, (int)N,                        (int)K, 0, splitk)
    if (M <= 32) LAUNCH_SK(2);
    else if (M <= 64) LAUNCH_SK(4);
    else LAUNCH_SK(8);
#undef LAUNCH_SK
    int64_t n = M * N;
    hipLaunchKernelGGL(skinny_epilogue_drop_kernel,
                       dim3(cdiv(n, 256)), dim3(256), 0, stream,
                       (const float*)yf.data_ptr(), (bf16*)y.data_ptr(),
                       (const int64_t*)seed.data_ptr(), n, splitk,
                       (float)p, (int)salt);
    HIP_OK(hipGetLastError());
    return y;
}

// ---- fused skinny epilogues for the BPTT step chain (B <= 32) ----

// gates epilogue + LSTM pointwise: thread per (b,h) sums the 4 gate
// elements across split-K slabs, applies TF LSTM gate math, writes the
// bf16 gates (saved for backward) plus h_raw and c_new.
__global__ void skinny_epi_lstm_kernel(const float* __restrict__ Yf,
                                       const bf16* __restrict__ bias,
                                       const bf16* __restrict__ c_prev,
                                       bf16* __restrict__ gates,
                                       bf16* __restrict__ h_out,
                                       bf16* __restrict__ c_out,
                                       int B, int H, int splitk,
                                       float fb) {
    int idx = blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= B * H) return;
    int b = idx / H, hh = idx % H;
    int64_t n = (int64_t)B * 4 * H;
    float g[4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
        int64_t off = (int64_t)b * 4 * H + q * H + hh;
        float acc = 0.f;
        for (int k = 0; k < splitk; ++k) acc += Yf[k * n + off];
        if (bias != nullptr) acc += bf2f(bias[q * H + hh]);
        gates[off] = f2bf(acc);
        g[q] = acc;
    }
    float gi = sigmoidf(g[0]);
    float gj = tanhf(g[1]);
    float gf = sigmoidf(g[2] + fb);
    float go = sigmoidf(g[3]);
    float cn = bf2f(c_prev[idx]) * gf + gi * gj;
    h_out[idx] = f2bf(tanhf(cn) * go);
    c_out[idx] = f2bf(cn);
}

std::vector<at::Tensor> dense_lstm_fwd(at::Tensor xh, at::Tensor wl,
                                       at::Tensor bl, at::Tensor c_prev,
                                       double fb) {
    // gates = xh @ wl^T + bl ; (h_raw, c_new) = LSTM(gates, c_prev)
    int64_t M = xh.size(0), K = xh.size(1), N = wl.size(0);
    int B = c_prev.size(0), H = c_prev.size(1);
    TORCH_CHECK(M == B && N == 4 * H && M <= 128 && K % 32 == 0);
    auto gates = at::empty({M, N}, xh.options());
    auto h_out = at::empty_like(c_prev);
    auto c_out = at::empty_like(c_prev);
    int nblocks = cdiv(N, 64);
    int splitk = 1;
    while (nblocks * splitk < 192 && splitk < 8 &&
           (int)(K / 32) >= 2 * splitk)
        splitk *= 2;
    auto yf = at::empty({splitk, M, N}, xh.options().dtype(at::kFloat));
    hipStream_t s = at::cuda::getCurrentCUDAStream();
#define LAUNCH_SLSTM(RF) \
    hipLaunchKernelGGL((skinny_gemm_kernel<RF>), dim3(nblocks, splitk), \
                       dim3(256), 0, s, (const bf16*)xh.data_ptr(), \
                       (const bf16*)wl.data_ptr(), nullptr, nullptr, \
                       (float*)yf.data_ptr(), nullptr, (int)M, (int)N, \
                       (int)K, ACT_NONE, splitk)
    if (M <= 32) LAUNCH_SLSTM(2);
    else if (M <= 64) LAUNCH_SLSTM(4);
    else LAUNCH_SLSTM(8);
#undef LAUNCH_SLSTM
    hipLaunchKernelGGL(skinny_epi_lstm_kernel,
                       dim3(cdiv(B * H, 256)), dim3(256), 0, s,
                       (const float*)yf.data_ptr(),
                       (const bf16*)bl.data_ptr(),
                       (const bf16*)c_prev.data_ptr(),
                       (bf16*)gates.data_ptr(), (bf16*)h_out.data_ptr(),
                       (bf16*)c_out.data_ptr(), B, H, splitk, (float)fb);
    HIP_OK(hipGetLastError());
    return {gates, h_out, c_out};
}

// epilogue + activation + hash dropout: writes both y and dropout(y)
__global__ void skinny_epi_drop_kernel(const float* __restrict__ Yf,
                                       const bf16* __restrict__ bias,
                                       const int64_t* __restrict__ seed_p,
                                       bf16* __restrict__ y,
                                       bf16* __restrict__ ydrop,
                                       int64_t n, int N, int act,
                                       int splitk, float p, int salt) {
    const uint32_t seed = (uint32_t)(*seed_p);
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    float v = 0.f;
    for (int k = 0; k < splitk; ++k) v += Yf[k * n + idx];
    if (bias != nullptr) v += bf2f(bias[idx % N]);
    v = apply_act(v, act);
    y[idx] = f2bf(v);
    float sc = 1.f;
    if (p > 0.f) {
        uint32_t h = (seed * 0x9E3779B1u) ^
                     ((uint32_t)salt * 0x85EBCA77u) ^
                     ((uint32_t)idx * 0xC2B2AE3Du);
        h ^= h >> 16; h *= 0x7FEB352Du;
        h ^= h >> 15; h *= 0x846CA68Bu;
        h ^= h >> 16;
        float u = (h >> 8) * (1.0f / 16777216.0f);
        sc = u >= p ? 1.0f / (1.0f - p) : 0.0f;
    }
    ydrop[idx] = f2bf(v * sc);
}

void dense_drop_fwd(at::Tensor x, at::Tensor w, at::Tensor b, int64_t act,
                    at::Tensor seed, double p, int64_t salt,
                    at::Tensor y, at::Tensor ydrop) {
    int64_t M = x.size(0), K = x.size(1), N = w.size(0);
    TORCH_CHECK(M <= 128 && K % 32 == 0);
    int nblocks = cdiv(N, 64);
    int splitk = 1;
    while (nblocks * splitk < 192 && splitk < 8 &&
           (int)(K / 32) >= 2 * splitk)
        splitk *= 2;
    auto yf = at::empty({splitk, M, N}, x.options().dtype(at::kFloat));
    hipStream_t s = at::cuda::getCurrentCUDAStream();
#define LAUNCH_SDROP(RF) \
    hipLaunchKernelGGL((skinny_gemm_kernel<RF>), dim3(nblocks, splitk), \
                       dim3(256), 0, s, (const bf16*)x.data_ptr(), \
                       (const bf16*)w.data_ptr(), nullptr, nullptr, \
                       (float*)yf.data_ptr(), nullptr, (int)M, (int)N, \
                       (int)K, ACT_NONE, splitk)
    if (M <= 32) LAUNCH_SDROP(2);
    else if (M <= 64) LAUNCH_SDROP(4);
    else LAUNCH_SDROP(8);
#undef LAUNCH_SDROP
    int64_t n = M * N;
    const bf16* bias_ptr = (b.defined() && b.numel() > 0)
        ? (const bf16*)b.data_ptr() : nullptr;
    hipLaunchKernelGGL(skinny_epi_drop_kernel, dim3(cdiv(n, 256)),
                       dim3(256), 0, s,
                       (const float*)yf.data_ptr(), bias_ptr,
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)y.data_ptr(), (bf16*)ydrop.data_ptr(),
                       n, (int)N, (int)act, splitk, (float)p, (int)salt);
    HIP_OK(hipGetLastError());
}

// ---------------------------------------------------------------------
// glds-staged 128x128 dense GEMM for big-M shapes (the attention
// projection T1: [T*B*L, 512] x [512, 512] ran at 148 TF on the
// register-staged 2-barrier tile; the glds + XOR-swizzle structure
// measured 536-658 TF on the conv igemm with identical geometry).
// Same discipline as conv_igemm_glds_kernel: global_load_lds width 16,
// lane-linear LDS, st_16x32 swizzle pre-applied to the per-lane GLOBAL
// source address and XOR'd on the fragment reads.  K % 64 == 0.
// ---------------------------------------------------------------------

__global__ __launch_bounds__(256)
void dense_glds_kernel(const bf16* __restrict__ A,   // [M,K]
                       const bf16* __restrict__ W,   // [N,K]
                       const bf16* __restrict__ bias,
                       bf16* __restrict__ Y,
                       int M, int N, int K, int act) {
    constexpr int BM = 128, BN = 128, BKc = 64;
    __shared__ bf16 lds[2 * BM * BKc];
    bf16* As = lds;
    bf16* Bs = lds + BM * BKc;

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = wave >> 1, wc = wave & 1;
    const int bm = blockIdx.y * BM;
    const int bn = blockIdx.x * BN;

    // per-lane staged rows (4 per wave for each operand) + source swz
    const bf16* aSrc[4];
    const bf16* wSrc[4];
    const int ci8 = (lane & 7) * 8;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int r = wave * 32 + i * 8 + (lane >> 3);
        int swz = (r & 4) ? 16 : 0;
        int ga = bm + r;
        if (ga >= M) ga = M - 1;
        aSrc[i] = A + (int64_t)ga * K + (ci8 ^ swz);
        int gb = bn + r;
        if (gb >= N) gb = N - 1;
        wSrc[i] = W + (int64_t)gb * K + (ci8 ^ swz);
    }

    floatx4 acc[4][4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
            acc[i][j] = floatx4{0.f, 0.f, 0.f, 0.f};

    const int lrow = lane & 15;
    const int kgrp = lane >> 4;

    for (int k0 = 0; k0 < K; k0 += BKc) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)
                    (aSrc[i] + k0),
                (__attribute__((address_space(3))) uint32_t*)
                    (As + (wave * 32 + i * 8) * BKc),
                16, 0, 0);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) uint32_t*)
                    (wSrc[i] + k0),
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
        float bv = (bias != nullptr && col < N) ? bf2f(bias[col]) : 0.f;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = bm + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
                if (row < M && col < N) {
                    float vv = acc[mi][ni][r] + bv;
                    Y[(int64_t)row * N + col] = f2bf(apply_act(vv, act));
                }
            }
        }
    }
}

// ---------------------------------------------------------------------
// dxh GEMM with the dx_fuse scatter folded into the split-K epilogue:
// dxh = dgates @ wl^T, then per element: input-dropout mask regen and
// the split into dpooled (+dpool_dec), DEMB slab (+demb_dec), dsth.
// Removes one launch + the dxh round trip per decoder reverse step.
// ---------------------------------------------------------------------

__global__ void skinny_epi_dx_kernel(const float* __restrict__ Yf,
                                     const bf16* __restrict__ dpool_dec,
                                     const bf16* __restrict__ demb_dec,
                                     const int64_t* __restrict__ seed_p,
                                     bf16* __restrict__ dpooled,
                                     bf16* __restrict__ demb_out,
                                     bf16* __restrict__ dsth,
                                     int64_t n, int splitk,
                                     int B, int D, int E, int H,
                                     float p, int salt) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int I = D + E;
    const int W = I + H;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    float g = 0.f;
    for (int k = 0; k < splitk; ++k) g += Yf[k * n + idx];
    int b = (int)(idx / W), j = (int)(idx % W);
    if (j < I) {
        g *= dscaleg(seed, salt, (uint32_t)(b * I + j), p);
        if (j < D)
            dpooled[(int64_t)b * D + j] =
                f2bf(g + bf2f(dpool_dec[(int64_t)b * D + j]));
        else
            demb_out[(int64_t)b * E + (j - D)] =
                f2bf(g + bf2f(demb_dec[(int64_t)b * E + (j - D)]));
    } else {
        dsth[(int64_t)b * H + (j - I)] = f2bf(g);
    }
}

std::vector<at::Tensor> dense_dx_fuse(at::Tensor dgates, at::Tensor wl_t,
                                      at::Tensor dpool_dec,
                                      at::Tensor demb_dec,
                                      at::Tensor seed,
                                      at::Tensor demb_out,
                                      double p, int64_t salt,
                                      int64_t D, int64_t E, int64_t H) {
    CHECK_GPU(dgates); CHECK_CONTIG(dgates); CHECK_BF16(dgates);
    CHECK_GPU(wl_t); CHECK_CONTIG(wl_t); CHECK_BF16(wl_t);
    int64_t M = dgates.size(0), K = dgates.size(1);
    int64_t N = wl_t.size(0);
    TORCH_CHECK(N == D + E + H);
    TORCH_CHECK(M <= 128 && K % 32 == 0,
                "dense_dx_fuse: skinny shapes only");
    auto dpooled = at::empty({M, D}, dgates.options());
    auto dsth = at::empty({M, H}, dgates.options());
    hipStream_t stream = at::cuda::getCurrentCUDAStream();

    int nblocks = cdiv(N, 64);
    int splitk = 1;
    while (nblocks * splitk < 192 && splitk < 8 &&
           (int)(K / 32) >= 2 * splitk)
        splitk *= 2;
    auto yf = at::empty({splitk, M, N}, dgates.options()
                                            .dtype(at::kFloat));
#define LAUNCH_SKX(RF) \
    hipLaunchKernelGGL((skinny_gemm_kernel<RF>), dim3(nblocks, splitk), \
                       dim3(256), 0, stream, \
                       (const bf16*)dgates.data_ptr(), \
                       (const bf16*)wl_t.data_ptr(), nullptr, nullptr, \
                       (float*)yf.data_ptr(), nullptr, (int)M, (int)N, \
                       (int)K, 0, splitk)
    if (M <= 32) LAUNCH_SKX(2);
    else if (M <= 64) LAUNCH_SKX(4);
    else LAUNCH_SKX(8);
#undef LAUNCH_SKX
    int64_t n = M * N;
    hipLaunchKernelGGL(skinny_epi_dx_kernel, dim3(cdiv(n, 256)),
                       dim3(256), 0, stream,
                       (const float*)yf.data_ptr(),
                       (const bf16*)dpool_dec.data_ptr(),
                       (const bf16*)demb_dec.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)dpooled.data_ptr(),
                       (bf16*)demb_out.data_ptr(),
                       (bf16*)dsth.data_ptr(),
                       n, splitk, (int)M, (int)D, (int)E, (int)H,
                       (float)p, (int)salt);
    HIP_OK(hipGetLastError());
    return {dpooled, dsth};
}

// ---------------------------------------------------------------------
// LSTM gates epilogue + the expand scatter in ONE launch: the H-range
// threads finish the gate math and write sth / EXPD-h / od_next
// directly (h_raw and out_t never touch HBM); D/E threads do the
// pooled / embedding-gather parts of the expand row.  Replaces
// skinny_epi_lstm + expand_fuse.  Dropout salts identical to the pair
// it replaces (s+4 out, s+5 state, s+6 expand, s+17 next attend).
// ---------------------------------------------------------------------

__global__ void skinny_epi_lstm_expand_kernel(
        const float* __restrict__ Yf, const bf16* __restrict__ bias,
        const bf16* __restrict__ c_prev,
        const bf16* __restrict__ pooled,
        const bf16* __restrict__ table,
        const int64_t* __restrict__ ids,
        const int64_t* __restrict__ seed_p,
        bf16* __restrict__ gates, bf16* __restrict__ c_out,
        bf16* __restrict__ sth_t, bf16* __restrict__ expdrop,
        bf16* __restrict__ od_next,
        int B, int H, int D, int E, int splitk,
        float fb, float p_lstm, float p_fc, int s) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int W = H + D + E;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)B * W) return;
    int b = idx / W, j = idx % W;
    float val;
    if (j < H) {
        int64_t n = (int64_t)B * 4 * H;
        float g[4];
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            int64_t off = (int64_t)b * 4 * H + q * H + j;
            float acc = 0.f;
            for (int k = 0; k < splitk; ++k) acc += Yf[k * n + off];
            if (bias != nullptr) acc += bf2f(bias[q * H + j]);
            gates[off] = f2bf(acc);
            g[q] = acc;
        }
        float gi = sigmoidf(g[0]);
        float gj = tanhf(g[1]);
        float gf = sigmoidf(g[2] + fb);
        float go = sigmoidf(g[3]);
        float cn = bf2f(c_prev[(int64_t)b * H + j]) * gf + gi * gj;
        float h = tanhf(cn) * go;
        c_out[(int64_t)b * H + j] = f2bf(cn);
        uint32_t hidx = (uint32_t)(b * H + j);
        float ot = h * dscaleg(seed, s + 4, hidx, p_lstm);
        sth_t[(int64_t)b * H + j] =
            f2bf(h * dscaleg(seed, s + 5, hidx, p_lstm));
        if (od_next != nullptr)
            od_next[(int64_t)b * H + j] =
                f2bf(ot * dscaleg(seed, s + 16 + 1, hidx, p_fc));
        val = ot;
    } else if (j < H + D) {
        val = bf2f(pooled[(int64_t)b * D + (j - H)]);
    } else {
        val = bf2f(table[ids[b] * E + (j - H - D)]);
    }
    expdrop[idx] = f2bf(val * dscaleg(seed, s + 6,
                                      (uint32_t)(b * W + j), p_fc));
}

std::vector<at::Tensor> dense_lstm_expand_fwd(
        at::Tensor xh, at::Tensor wl, at::Tensor bl, at::Tensor c_prev,
        at::Tensor pooled, at::Tensor table, at::Tensor ids,
        at::Tensor seed, at::Tensor expdrop, at::Tensor od_next,
        double fb, double p_lstm, double p_fc, int64_t s) {
    int64_t M = xh.size(0), K = xh.size(1), N = wl.size(0);
    int B = c_prev.size(0), H = c_prev.size(1);
    int D = pooled.size(1), E = table.size(1);
    TORCH_CHECK(M == B && N == 4 * H && M <= 128 && K % 32 == 0);
    auto gates = at::empty({M, N}, xh.options());
    auto c_out = at::empty({B, H}, xh.options());
    auto sth_t = at::empty({B, H}, xh.options());
    hipStream_t stream = at::cuda::getCurrentCUDAStream();

    int nblocks = cdiv(N, 64);
    int splitk = 1;
    while (nblocks * splitk < 192 && splitk < 8 &&
           (int)(K / 32) >= 2 * splitk)
        splitk *= 2;
    auto yf = at::empty({splitk, M, N},
                        xh.options().dtype(at::kFloat));
#define LAUNCH_SKL(RF) \
    hipLaunchKernelGGL((skinny_gemm_kernel<RF>), dim3(nblocks, splitk), \
                       dim3(256), 0, stream, \
                       (const bf16*)xh.data_ptr(), \
                       (const bf16*)wl.data_ptr(), nullptr, nullptr, \
                       (float*)yf.data_ptr(), nullptr, (int)M, (int)N, \
                       (int)K, 0, splitk)
    if (M <= 32) LAUNCH_SKL(2);
    else if (M <= 64) LAUNCH_SKL(4);
    else LAUNCH_SKL(8);
#undef LAUNCH_SKL
    const bf16* bias_ptr = nullptr;
    if (bl.defined() && bl.numel() > 0)
        bias_ptr = (const bf16*)bl.data_ptr();
    bf16* od_ptr = nullptr;
    if (od_next.defined() && od_next.numel() > 0)
        od_ptr = (bf16*)od_next.data_ptr();
    int64_t n = (int64_t)B * (H + D + E);
    hipLaunchKernelGGL(skinny_epi_lstm_expand_kernel,
                       dim3(cdiv(n, 256)), dim3(256), 0, stream,
                       (const float*)yf.data_ptr(), bias_ptr,
                       (const bf16*)c_prev.data_ptr(),
                       (const bf16*)pooled.data_ptr(),
                       (const bf16*)table.data_ptr(),
                       (const int64_t*)ids.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)gates.data_ptr(),
                       (bf16*)c_out.data_ptr(),
                       (bf16*)sth_t.data_ptr(),
                       (bf16*)expdrop.data_ptr(), od_ptr,
                       B, H, D, E, splitk,
                       (float)fb, (float)p_lstm, (float)p_fc, (int)s);
    HIP_OK(hipGetLastError());
    return {gates, c_out, sth_t};
}
