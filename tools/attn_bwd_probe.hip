// Standalone microbench: isolate why attn_scores_bwd is 40x slower than
// its traffic floor.  hipcc --offload-arch=gfx950 -O3 attn_bwd_probe.hip
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

#define B 32
#define L 196
#define A 512

__device__ __forceinline__ float wsum(float v) {
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    return v;
}

// v1: current structure (wave-per-row, atomics at end)
template <int ATOMICS>
__global__ void bwd_v1(const bf16* tdrop, const bf16* v,
                       const float* dlogits, bf16* dt1, float* dt2,
                       float* dvf, int lchunk) {
    int nchunk = (L + lchunk - 1) / lchunk;
    int b = blockIdx.x / nchunk;
    int l0 = (blockIdx.x % nchunk) * lchunk;
    int l1 = min(L, l0 + lchunk);
    int wid = threadIdx.x >> 6, lane = threadIdx.x & 63;
    float dv_acc[8] = {};
    float dt2_acc[8] = {};
    for (int l = l0 + wid; l < l1; l += 4) {
        long row = (long)b * L + l;
        float dl = dlogits[row];
        int a0 = lane * 8;
        bf16x8 td = *(const bf16x8*)(tdrop + row * A + a0);
        bf16x8 vv = *(const bf16x8*)(v + a0);
        bf16x8 o;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
            dv_acc[e] += (float)td[e] * dl;
            float dt = dl * (float)vv[e];
            o[e] = (bf16)dt;
            dt2_acc[e] += dt;
        }
        *(bf16x8*)(dt1 + row * A + a0) = o;
    }
    int a0 = lane * 8;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        if (ATOMICS) {
            atomicAdd(dvf + a0 + e, dv_acc[e]);
            atomicAdd(dt2 + (long)b * A + a0 + e, dt2_acc[e]);
        } else {
            dvf[a0 + e] = dv_acc[e];
            dt2[(long)b * A + a0 + e] = dt2_acc[e];
        }
    }
}

// v2: thread-per-column, block sweeps rows sequentially
template <int ATOMICS>
__global__ void bwd_v2(const bf16* tdrop, const bf16* v,
                       const float* dlogits, bf16* dt1, float* dt2,
                       float* dvf, int lchunk) {
    int nchunk = (L + lchunk - 1) / lchunk;
    int b = blockIdx.x / nchunk;
    int l0 = (blockIdx.x % nchunk) * lchunk;
    int l1 = min(L, l0 + lchunk);
    int tid = threadIdx.x;
    float dv0 = 0, dv1 = 0, d20 = 0, d21 = 0;
    float v0 = (float)v[tid], v1 = (float)v[tid + 256];
    for (int l = l0; l < l1; ++l) {
        long row = (long)b * L + l;
        float dl = dlogits[row];
        float t0 = (float)tdrop[row * A + tid];
        float t1 = (float)tdrop[row * A + tid + 256];
        dv0 += t0 * dl; dv1 += t1 * dl;
        float a = dl * v0, c = dl * v1;
        dt1[row * A + tid] = (bf16)a;
        dt1[row * A + tid + 256] = (bf16)c;
        d20 += a; d21 += c;
    }
    if (ATOMICS) {
        atomicAdd(dvf + tid, dv0);
        atomicAdd(dvf + tid + 256, dv1);
        atomicAdd(dt2 + (long)b * A + tid, d20);
        atomicAdd(dt2 + (long)b * A + tid + 256, d21);
    } else {
        dvf[tid] = dv0; dvf[tid + 256] = dv1;
        dt2[(long)b * A + tid] = d20;
        dt2[(long)b * A + tid + 256] = d21;
    }
}

template <typename F>
float timeit(F f, int iters) {
    for (int i = 0; i < 20; ++i) f();
    hipDeviceSynchronize();
    hipEvent_t s, e;
    hipEventCreate(&s); hipEventCreate(&e);
    hipEventRecord(s);
    for (int i = 0; i < iters; ++i) f();
    hipEventRecord(e);
    hipEventSynchronize(e);
    float ms;
    hipEventElapsedTime(&ms, s, e);
    return ms / iters * 1000.f;
}

int main() {
    bf16 *tdrop, *v, *dt1;
    float *dlogits, *dt2, *dvf;
    hipMalloc(&tdrop, (long)B * L * A * 2);
    hipMalloc(&v, A * 2);
    hipMalloc(&dt1, (long)B * L * A * 2);
    hipMalloc(&dlogits, (long)B * L * 4);
    hipMalloc(&dt2, (long)B * A * 4);
    hipMalloc(&dvf, A * 4);
    hipMemset(tdrop, 0x3c, (long)B * L * A * 2);
    hipMemset(dlogits, 0, (long)B * L * 4);

    for (int lchunk : {49, 13}) {
        int nblocks = B * ((L + lchunk - 1) / lchunk);
        printf("lchunk=%d blocks=%d\n", lchunk, nblocks);
        printf("  v1 atomics:    %7.1f us\n",
               timeit([&] { hipLaunchKernelGGL((bwd_v1<1>), dim3(nblocks),
                            dim3(256), 0, 0, tdrop, v, dlogits, dt1, dt2,
                            dvf, lchunk); }, 100));
        printf("  v1 no-atomics: %7.1f us\n",
               timeit([&] { hipLaunchKernelGGL((bwd_v1<0>), dim3(nblocks),
                            dim3(256), 0, 0, tdrop, v, dlogits, dt1, dt2,
                            dvf, lchunk); }, 100));
        printf("  v2 atomics:    %7.1f us\n",
               timeit([&] { hipLaunchKernelGGL((bwd_v2<1>), dim3(nblocks),
                            dim3(256), 0, 0, tdrop, v, dlogits, dt1, dt2,
                            dvf, lchunk); }, 100));
        printf("  v2 no-atomics: %7.1f us\n",
               timeit([&] { hipLaunchKernelGGL((bwd_v2<0>), dim3(nblocks),
                            dim3(256), 0, 0, tdrop, v, dlogits, dt1, dt2,
                            dvf, lchunk); }, 100));
    }
    return 0;
}
