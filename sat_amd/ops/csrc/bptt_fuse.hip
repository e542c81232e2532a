// Per-step pointwise chain fusions for the hand-written BPTT
// (sat_amd/models/bptt.py).  Each kernel replaces a 4-8 launch eager chain
// (cat / slice / dropout / add) with one launch writing straight into the
// step-major batched buffers.  Dropout masks use the same counter hash as
// kernels.hip (seed value hoisted once per kernel), with flat per-tensor
// indices identical to the standalone hash_dropout calls they replace.

#include "common.h"

__device__ __forceinline__ uint32_t mix3b(uint32_t a, uint32_t b,
                                          uint32_t c) {
    uint32_t h = a * 0x9E3779B1u ^ b * 0x85EBCA77u ^ c * 0xC2B2AE3Du;
    h ^= h >> 16; h *= 0x7FEB352Du;
    h ^= h >> 15; h *= 0x846CA68Bu;
    h ^= h >> 16;
    return h;
}

__device__ __forceinline__ float dscale(uint32_t seed, int salt,
                                        uint32_t idx, float p) {
    if (p <= 0.f) return 1.f;
    uint32_t h = mix3b(seed, (uint32_t)salt, idx);
    float u = (h >> 8) * (1.0f / 16777216.0f);
    return u >= p ? 1.0f / (1.0f - p) : 0.0f;
}

// ---- forward: xh = [dropout(cat(pooled, emb), p_lstm, salt), state_h] ----

__global__ void lstm_in_fuse_kernel(const bf16* __restrict__ pooled,
                                    const bf16* __restrict__ table,
                                    const int64_t* __restrict__ ids,
                                    const bf16* __restrict__ sth,
                                    const int64_t* __restrict__ seed_p,
                                    bf16* __restrict__ xh,
                                    int B, int D, int E, int H,
                                    float p, int salt) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int I = D + E;
    const int W = I + H;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)B * W) return;
    int b = idx / W, j = idx % W;
    float val;
    if (j < I) {
        val = (j < D) ? bf2f(pooled[(int64_t)b * D + j])
                      : bf2f(table[ids[b] * E + (j - D)]);
        val *= dscale(seed, salt, (uint32_t)(b * I + j), p);
    } else {
        val = bf2f(sth[(int64_t)b * H + (j - I)]);
    }
    xh[idx] = f2bf(val);
}

void lstm_in_fuse(at::Tensor pooled, at::Tensor table, at::Tensor ids,
                  at::Tensor sth, at::Tensor seed, double p, int64_t salt,
                  at::Tensor xh) {
    int B = pooled.size(0), D = pooled.size(1), E = table.size(1),
        H = sth.size(1);
    int64_t n = (int64_t)B * (D + E + H);
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(lstm_in_fuse_kernel, dim3(cdiv(n, 256)), dim3(256),
                       0, s,
                       (const bf16*)pooled.data_ptr(),
                       (const bf16*)table.data_ptr(),
                       (const int64_t*)ids.data_ptr(),
                       (const bf16*)sth.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)xh.data_ptr(), B, D, E, H,
                       (float)p, (int)salt);
    HIP_OK(hipGetLastError());
}

// ---- forward: out/state dropouts + expanded concat + dropout, plus the
// NEXT step's attend-input dropout of out_t (salt+16+1) ----

__global__ void expand_fuse_kernel(const bf16* __restrict__ h_raw,
                                   const bf16* __restrict__ pooled,
                                   const bf16* __restrict__ table,
                                   const int64_t* __restrict__ ids,
                                   const int64_t* __restrict__ seed_p,
                                   bf16* __restrict__ out_t,
                                   bf16* __restrict__ sth_t,
                                   bf16* __restrict__ expdrop,
                                   bf16* __restrict__ od_next,
                                   int B, int H, int D, int E,
                                   float p_lstm, float p_fc, int s) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int W = H + D + E;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)B * W) return;
    int b = idx / W, j = idx % W;
    float val;
    if (j < H) {
        float h = bf2f(h_raw[(int64_t)b * H + j]);
        uint32_t hidx = (uint32_t)(b * H + j);
        float ot = h * dscale(seed, s + 4, hidx, p_lstm);
        out_t[(int64_t)b * H + j] = f2bf(ot);
        sth_t[(int64_t)b * H + j] =
            f2bf(h * dscale(seed, s + 5, hidx, p_lstm));
        if (od_next != nullptr)
            od_next[(int64_t)b * H + j] =
                f2bf(ot * dscale(seed, s + 16 + 1, hidx, p_fc));
        val = ot;
    } else if (j < H + D) {
        val = bf2f(pooled[(int64_t)b * D + (j - H)]);
    } else {
        val = bf2f(table[ids[b] * E + (j - H - D)]);
    }
    expdrop[idx] = f2bf(val * dscale(seed, s + 6, (uint32_t)(b * W + j),
                                     p_fc));
}

std::vector<at::Tensor> expand_fuse(at::Tensor h_raw, at::Tensor pooled,
                                    at::Tensor table, at::Tensor ids,
                                    at::Tensor seed,
                                    at::Tensor expdrop, at::Tensor od_next,
                                    double p_lstm, double p_fc,
                                    int64_t s) {
    int B = h_raw.size(0), H = h_raw.size(1), D = pooled.size(1),
        E = table.size(1);
    auto out_t = at::empty_like(h_raw);
    auto sth_t = at::empty_like(h_raw);
    bf16* od_ptr = nullptr;
    if (od_next.defined() && od_next.numel() > 0)
        od_ptr = (bf16*)od_next.data_ptr();
    int64_t n = (int64_t)B * (H + D + E);
    hipStream_t st = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(expand_fuse_kernel, dim3(cdiv(n, 256)), dim3(256),
                       0, st,
                       (const bf16*)h_raw.data_ptr(),
                       (const bf16*)pooled.data_ptr(),
                       (const bf16*)table.data_ptr(),
                       (const int64_t*)ids.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)out_t.data_ptr(), (bf16*)sth_t.data_ptr(),
                       (bf16*)expdrop.data_ptr(), od_ptr,
                       B, H, D, E, (float)p_lstm, (float)p_fc, (int)s);
    HIP_OK(hipGetLastError());
    return {out_t, sth_t};
}

// ---- backward: dexpanded -> (dh_raw with carries, dpool_dec, demb_dec) ----

__global__ void dexp_fuse_kernel(const bf16* __restrict__ dexpd,
                                 const bf16* __restrict__ d_out_carry,
                                 const bf16* __restrict__ d_sth_carry,
                                 const int64_t* __restrict__ seed_p,
                                 bf16* __restrict__ dh_raw,
                                 bf16* __restrict__ dpool_dec,
                                 bf16* __restrict__ demb_dec,
                                 int B, int H, int D, int E,
                                 float p_fc, float p_lstm, int s) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int W = H + D + E;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)B * W) return;
    int b = idx / W, j = idx % W;
    float dexp = bf2f(dexpd[idx])
        * dscale(seed, s + 6, (uint32_t)(b * W + j), p_fc);
    if (j < H) {
        uint32_t hidx = (uint32_t)(b * H + j);
        float dout = dexp + bf2f(d_out_carry[(int64_t)b * H + j]);
        float dh = dout * dscale(seed, s + 4, hidx, p_lstm)
            + bf2f(d_sth_carry[(int64_t)b * H + j])
            * dscale(seed, s + 5, hidx, p_lstm);
        dh_raw[(int64_t)b * H + j] = f2bf(dh);
    } else if (j < H + D) {
        dpool_dec[(int64_t)b * D + (j - H)] = f2bf(dexp);
    } else {
        demb_dec[(int64_t)b * E + (j - H - D)] = f2bf(dexp);
    }
}

std::vector<at::Tensor> dexp_fuse(at::Tensor dexpd, at::Tensor d_out_carry,
                                  at::Tensor d_sth_carry, at::Tensor seed,
                                  double p_fc, double p_lstm, int64_t s,
                                  int64_t D, int64_t E) {
    int B = d_out_carry.size(0), H = d_out_carry.size(1);
    auto dh_raw = at::empty_like(d_out_carry);
    auto dpool = at::empty({B, D}, d_out_carry.options());
    auto demb = at::empty({B, E}, d_out_carry.options());
    int64_t n = (int64_t)B * (H + D + E);
    hipStream_t st = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dexp_fuse_kernel, dim3(cdiv(n, 256)), dim3(256),
                       0, st,
                       (const bf16*)dexpd.data_ptr(),
                       (const bf16*)d_out_carry.data_ptr(),
                       (const bf16*)d_sth_carry.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)dh_raw.data_ptr(), (bf16*)dpool.data_ptr(),
                       (bf16*)demb.data_ptr(), B, H, (int)D, (int)E,
                       (float)p_fc, (float)p_lstm, (int)s);
    HIP_OK(hipGetLastError());
    return {dh_raw, dpool, demb};
}

// ---- backward: dxh -> (dpooled, demb written to buffer, d_sth_carry) ----

__global__ void dx_fuse_kernel(const bf16* __restrict__ dxh,
                               const bf16* __restrict__ dpool_dec,
                               const bf16* __restrict__ demb_dec,
                               const int64_t* __restrict__ seed_p,
                               bf16* __restrict__ dpooled,
                               bf16* __restrict__ demb_out,
                               bf16* __restrict__ dsth,
                               int B, int D, int E, int H,
                               float p, int salt) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int I = D + E;
    const int W = I + H;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)B * W) return;
    int b = idx / W, j = idx % W;
    float g = bf2f(dxh[idx]);
    if (j < I) {
        g *= dscale(seed, salt, (uint32_t)(b * I + j), p);
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

std::vector<at::Tensor> dx_fuse(at::Tensor dxh, at::Tensor dpool_dec,
                                at::Tensor demb_dec, at::Tensor seed,
                                at::Tensor demb_out, double p,
                                int64_t salt, int64_t H) {
    int B = dpool_dec.size(0), D = dpool_dec.size(1), E = demb_dec.size(1);
    auto dpooled = at::empty_like(dpool_dec);
    auto dsth = at::empty({B, H}, dpool_dec.options());
    int64_t n = (int64_t)B * (D + E + H);
    hipStream_t st = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dx_fuse_kernel, dim3(cdiv(n, 256)), dim3(256),
                       0, st,
                       (const bf16*)dxh.data_ptr(),
                       (const bf16*)dpool_dec.data_ptr(),
                       (const bf16*)demb_dec.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)dpooled.data_ptr(),
                       (bf16*)demb_out.data_ptr(),
                       (bf16*)dsth.data_ptr(),
                       B, D, E, (int)H, (float)p, (int)salt);
    HIP_OK(hipGetLastError());
    return {dpooled, dsth};
}

// ---- hash_dropout variant writing into a caller-provided buffer ----

__global__ void hash_dropout_out_kernel(const bf16* __restrict__ x,
                                        const int64_t* __restrict__ seed_p,
                                        bf16* __restrict__ y,
                                        int64_t n, float p, int salt) {
    const uint32_t seed = (uint32_t)(*seed_p);
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    y[idx] = f2bf(bf2f(x[idx]) * dscale(seed, salt, (uint32_t)idx, p));
}

void hash_dropout_out(at::Tensor x, at::Tensor seed, double p,
                      int64_t salt, at::Tensor out) {
    int64_t n = x.numel();
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(hash_dropout_out_kernel, dim3(cdiv(n, 256)),
                       dim3(256), 0, s,
                       (const bf16*)x.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)out.data_ptr(), n, (float)p, (int)salt);
    HIP_OK(hipGetLastError());
}

// ---- generate all T step-dropouts of one tensor into a [T, n] buffer ----
// slab t uses salt = salt_base + t*salt_stride with slab-local indices,
// exactly matching T separate hash_dropout calls.

__global__ void hash_dropout_steps_kernel(const bf16* __restrict__ x,
                                          const int64_t* __restrict__ seed_p,
                                          bf16* __restrict__ y,
                                          int64_t n, int T, float p,
                                          int salt_base, int salt_stride) {
    // n % 8 == 0 (checked by the wrapper): bf16x8 vector traffic
    const uint32_t seed = (uint32_t)(*seed_p);
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    int64_t total = n * T;
    int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
    for (; i8 < total; i8 += stride) {
        int t = (int)(i8 / n);
        int64_t j = i8 % n;
        int salt = salt_base + t * salt_stride;
        bf16x8 v = *(const bf16x8*)(x + j);
        bf16x8 o;
#pragma unroll
        for (int e = 0; e < 8; ++e)
            o[e] = f2bf(bf2f(v[e]) * dscale(seed, salt,
                                            (uint32_t)(j + e), p));
        *(bf16x8*)(y + i8) = o;
    }
}

at::Tensor hash_dropout_steps(at::Tensor x, at::Tensor seed, double p,
                              int64_t salt_base, int64_t salt_stride,
                              int64_t T) {
    int64_t n = x.numel();
    TORCH_CHECK(n % 8 == 0, "hash_dropout_steps needs numel % 8 == 0");
    auto y = at::empty({T, x.size(0), x.size(1)}, x.options());
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    int blocks = (int)std::min<int64_t>(cdiv(n * T, 256 * 8), 16384);
    hipLaunchKernelGGL(hash_dropout_steps_kernel, dim3(blocks), dim3(256),
                       0, s,
                       (const bf16*)x.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)y.data_ptr(), n, (int)T, (float)p,
                       (int)salt_base, (int)salt_stride);
    HIP_OK(hipGetLastError());
    return y;
}

__global__ void act_bwd_out_kernel(const bf16* __restrict__ dy,
                                   const bf16* __restrict__ y,
                                   bf16* __restrict__ dpre,
                                   int64_t n, int act) {
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    float g = bf2f(dy[idx]);
    float yv = bf2f(y[idx]);
    if (act == 1) g *= (1.f - yv * yv);
    else if (act == 2) g *= (yv > 0.f ? 1.f : 0.f);
    dpre[idx] = f2bf(g);
}

void act_bwd_out(at::Tensor dy, at::Tensor y, int64_t act, at::Tensor out) {
    int64_t n = dy.numel();
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(act_bwd_out_kernel, dim3(cdiv(n, 256)), dim3(256),
                       0, s,
                       (const bf16*)dy.data_ptr(), (const bf16*)y.data_ptr(),
                       (bf16*)out.data_ptr(), n, (int)act);
    HIP_OK(hipGetLastError());
}

// fp32-dy variant writing into a preallocated bf16 slab: replaces the
// per-step  dt2f.to(bf16) -> act_bwd -> DPRE1B[sl].copy_  three-kernel
// chain in the core BPTT backward with ONE launch
__global__ void act_bwd_f32_out_kernel(const float* __restrict__ dy,
                                       const bf16* __restrict__ y,
                                       bf16* __restrict__ dpre,
                                       int64_t n, int act) {
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= n) return;
    float g = dy[idx];
    float yv = bf2f(y[idx]);
    if (act == 1) g *= (1.f - yv * yv);
    else if (act == 2) g *= (yv > 0.f ? 1.f : 0.f);
    dpre[idx] = f2bf(g);
}

void act_bwd_f32_out(at::Tensor dy, at::Tensor y, int64_t act,
                     at::Tensor out) {
    int64_t n = dy.numel();
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(act_bwd_f32_out_kernel, dim3(cdiv(n, 256)),
                       dim3(256), 0, s,
                       (const float*)dy.data_ptr(),
                       (const bf16*)y.data_ptr(),
                       (bf16*)out.data_ptr(), n, (int)act);
    HIP_OK(hipGetLastError());
}

// ---- dropout over a step-major [T*B, N] buffer where slab t uses salt
// salt_base + t*salt_stride and slab-local flat indices (matches the
// per-step hash_dropout calls it replaces) ----

__global__ void hash_dropout_slabs_kernel(const bf16* __restrict__ x,
                                          const int64_t* __restrict__ seed_p,
                                          bf16* __restrict__ y,
                                          int64_t slab_elems, int T,
                                          float p, int salt_base,
                                          int salt_stride) {
    const uint32_t seed = (uint32_t)(*seed_p);
    int64_t i8 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    int64_t total = slab_elems * T;
    if (i8 >= total) return;
    int t = (int)(i8 / slab_elems);
    int64_t j = i8 % slab_elems;
    int salt = salt_base + t * salt_stride;
    bf16x8 v = *(const bf16x8*)(x + i8);
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e)
        o[e] = f2bf(bf2f(v[e]) * dscale(seed, salt, (uint32_t)(j + e), p));
    *(bf16x8*)(y + i8) = o;
}

at::Tensor hash_dropout_slabs(at::Tensor x, at::Tensor seed, double p,
                              int64_t salt_base, int64_t salt_stride,
                              int64_t T) {
    if (p <= 0.0) return x;
    int64_t n = x.numel();
    int64_t slab = n / T;
    TORCH_CHECK(slab % 8 == 0 && n % T == 0);
    auto y = at::empty_like(x);
    hipStream_t s = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(hash_dropout_slabs_kernel,
                       dim3(cdiv(n / 8, 256)), dim3(256), 0, s,
                       (const bf16*)x.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (bf16*)y.data_ptr(), slab, (int)T, (float)p,
                       (int)salt_base, (int)salt_stride);
    HIP_OK(hipGetLastError());
    return y;
}

// ---- dexp_fuse + LSTM pointwise backward in ONE launch: the H-range
// threads continue straight into the gate math (dh_raw never touches
// HBM); D/E-range threads do the dexp scatter as before ----

__global__ void dexp_lstm_bwd_kernel(const bf16* __restrict__ dexpd,
                                     const bf16* __restrict__ d_out_carry,
                                     const bf16* __restrict__ d_sth_carry,
                                     const int64_t* __restrict__ seed_p,
                                     const bf16* __restrict__ gates,
                                     const bf16* __restrict__ c,
                                     const bf16* __restrict__ dc,
                                     bf16* __restrict__ dgates,
                                     bf16* __restrict__ dc_prev,
                                     bf16* __restrict__ dpool_dec,
                                     bf16* __restrict__ demb_dec,
                                     int B, int H, int D, int E,
                                     float p_fc, float p_lstm, int s,
                                     float fb) {
    const uint32_t seed = (uint32_t)(*seed_p);
    const int W = H + D + E;
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= (int64_t)B * W) return;
    int b = idx / W, j = idx % W;
    float dexp = bf2f(dexpd[idx])
        * dscale(seed, s + 6, (uint32_t)(b * W + j), p_fc);
    if (j < H) {
        uint32_t hidx = (uint32_t)(b * H + j);
        float dout = dexp + bf2f(d_out_carry[(int64_t)b * H + j]);
        float dhv = dout * dscale(seed, s + 4, hidx, p_lstm)
            + bf2f(d_sth_carry[(int64_t)b * H + j])
            * dscale(seed, s + 5, hidx, p_lstm);
        // LSTM pointwise backward (same math as lstm_pw_bwd_kernel)
        const bf16* g = gates + (int64_t)b * 4 * H;
        float gi = 1.f / (1.f + __expf(-bf2f(g[j])));
        float gj = tanhf(bf2f(g[H + j]));
        float gf = 1.f / (1.f + __expf(-(bf2f(g[2 * H + j]) + fb)));
        float go = 1.f / (1.f + __expf(-bf2f(g[3 * H + j])));
        float cp = bf2f(c[(int64_t)b * H + j]);
        float cn = cp * gf + gi * gj;
        float tc = tanhf(cn);
        float dcv = (dc != nullptr) ? bf2f(dc[(int64_t)b * H + j]) : 0.f;
        float dct = dcv + dhv * go * (1.f - tc * tc);
        bf16* dg = dgates + (int64_t)b * 4 * H;
        dg[j]         = f2bf(dct * gj * gi * (1.f - gi));
        dg[H + j]     = f2bf(dct * gi * (1.f - gj * gj));
        dg[2 * H + j] = f2bf(dct * cp * gf * (1.f - gf));
        dg[3 * H + j] = f2bf(dhv * tc * go * (1.f - go));
        dc_prev[(int64_t)b * H + j] = f2bf(dct * gf);
    } else if (j < H + D) {
        dpool_dec[(int64_t)b * D + (j - H)] = f2bf(dexp);
    } else {
        demb_dec[(int64_t)b * E + (j - H - D)] = f2bf(dexp);
    }
}

std::vector<at::Tensor> dexp_lstm_bwd(at::Tensor dexpd,
                                      at::Tensor d_out_carry,
                                      at::Tensor d_sth_carry,
                                      at::Tensor seed, at::Tensor gates,
                                      at::Tensor c_prev, at::Tensor dc,
                                      at::Tensor dgates_out,
                                      double p_fc, double p_lstm,
                                      int64_t s, int64_t D, int64_t E,
                                      double fb) {
    CHECK_GPU(dexpd); CHECK_BF16(dexpd);
    int B = d_out_carry.size(0), H = d_out_carry.size(1);
    TORCH_CHECK(dexpd.numel() == (int64_t)B * (H + D + E));
    TORCH_CHECK(dgates_out.numel() == (int64_t)B * 4 * H);
    auto dc_prev = at::empty({B, H}, dexpd.options());
    auto dpool_dec = at::empty({B, D}, dexpd.options());
    auto demb_dec = at::empty({B, E}, dexpd.options());
    const bf16* dc_ptr = nullptr;
    if (dc.defined() && dc.numel() > 0)
        dc_ptr = (const bf16*)dc.data_ptr();
    int64_t n = (int64_t)B * (H + D + E);
    hipStream_t st = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(dexp_lstm_bwd_kernel, dim3(cdiv(n, 256)),
                       dim3(256), 0, st,
                       (const bf16*)dexpd.data_ptr(),
                       (const bf16*)d_out_carry.data_ptr(),
                       (const bf16*)d_sth_carry.data_ptr(),
                       (const int64_t*)seed.data_ptr(),
                       (const bf16*)gates.data_ptr(),
                       (const bf16*)c_prev.data_ptr(), dc_ptr,
                       (bf16*)dgates_out.data_ptr(),
                       (bf16*)dc_prev.data_ptr(),
                       (bf16*)dpool_dec.data_ptr(),
                       (bf16*)demb_dec.data_ptr(),
                       B, H, (int)D, (int)E,
                       (float)p_fc, (float)p_lstm, (int)s, (float)fb);
    HIP_OK(hipGetLastError());
    return {dc_prev, dpool_dec, demb_dec};
}
