// Shared helpers for pipegoose_amd CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

// ---------------------------------------------------------------- conversions
__device__ __forceinline__ float to_float(float v) { return v; }
__device__ __forceinline__ float to_float(__hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T>
__device__ __forceinline__ T from_float(float v);
template <>
__device__ __forceinline__ float from_float<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_float<__hip_bfloat16>(float v) {
    return __float2bfloat16(v);
}

// ------------------------------------------------------- wave/block reductions
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
        v += __shfl_down(v, off, WAVE_SIZE);
    }
    return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) {
        v = fmaxf(v, __shfl_down(v, off, WAVE_SIZE));
    }
    return v;
}

// Block-level sum; every thread returns the result. `smem` needs
// blockDim.x/WAVE_SIZE floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* smem) {
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x / WAVE_SIZE;
    const int n_waves = blockDim.x / WAVE_SIZE;
    v = wave_reduce_sum(v);
    if (lane == 0) smem[wave] = v;
    __syncthreads();
    v = (threadIdx.x < n_waves) ? smem[threadIdx.x] : 0.0f;
    if (wave == 0) {
        v = wave_reduce_sum(v);
        if (lane == 0) smem[0] = v;
    }
    __syncthreads();
    v = smem[0];
    __syncthreads();
    return v;
}

__device__ __forceinline__ float block_reduce_max(float v, float* smem) {
    const int lane = threadIdx.x & (WAVE_SIZE - 1);
    const int wave = threadIdx.x / WAVE_SIZE;
    const int n_waves = blockDim.x / WAVE_SIZE;
    v = wave_reduce_max(v);
    if (lane == 0) smem[wave] = v;
    __syncthreads();
    v = (threadIdx.x < n_waves) ? smem[threadIdx.x] : -INFINITY;
    if (wave == 0) {
        v = wave_reduce_max(v);
        if (lane == 0) smem[0] = v;
    }
    __syncthreads();
    v = smem[0];
    __syncthreads();
    return v;
}

// ------------------------------------------------------------- vector packets
// 8 consecutive bf16 (16 B) or 8 consecutive fp32 (2x16 B): the coalescing
// sweet spot per the CDNA4 guide (G13).
struct bf16x8 {
    __hip_bfloat16 v[8];
};
struct f32x8 {
    float v[8];
};

template <typename T>
struct vec8;
template <>
struct vec8<__hip_bfloat16> {
    using type = bf16x8;
};
template <>
struct vec8<float> {
    using type = f32x8;
};

#define HIP_CHECK_LAUNCH()                                             \
    do {                                                               \
        hipError_t e_ = hipGetLastError();                             \
        if (e_ != hipSuccess) {                                        \
            TORCH_CHECK(false, "HIP kernel launch failed: ",           \
                        hipGetErrorString(e_));                        \
        }                                                              \
    } while (0)
