// Common helpers for the sat_amd CDNA4 (gfx950) kernel layer.
// MI355X-only: wave64, MFMA bf16, 160 KiB LDS/CU. No CUDA paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")
#define CHECK_CONTIG(x) TORCH_CHECK((x).is_contiguous(), #x " must be contiguous")
#define CHECK_BF16(x) TORCH_CHECK((x).scalar_type() == at::kBFloat16, #x " must be bf16")
#define CHECK_F32(x) TORCH_CHECK((x).scalar_type() == at::kFloat, #x " must be fp32")

#define HIP_OK(expr) do { hipError_t _e = (expr); \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); } while (0)

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float floatx4;

static inline int cdiv(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

__device__ __forceinline__ float bf2f(bf16 v) { return (float)v; }
__device__ __forceinline__ bf16 f2bf(float v) { return (bf16)v; }

// wave-level sum over 64 lanes
__device__ __forceinline__ float wave_sum(float v) {
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return v;
}

__device__ __forceinline__ float wave_max(float v) {
    for (int off = 32; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_down(v, off, 64));
    return v;
}

__device__ __forceinline__ float sigmoidf(float x) {
    return 1.0f / (1.0f + __expf(-x));
}
