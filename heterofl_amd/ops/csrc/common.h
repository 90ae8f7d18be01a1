// Shared helpers for the gfx950 HIP kernels (CDNA4: wave64, 4xSIMD-32 CUs).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

// fp32 accumulate load for fp32 / bf16 tensors
template <typename T>
__device__ __forceinline__ float ld_f32(const T* p) { return (float)*p; }
template <>
__device__ __forceinline__ float ld_f32<__hip_bfloat16>(const __hip_bfloat16* p) {
    return __bfloat162float(*p);
}

template <typename T>
__device__ __forceinline__ void st_f32(T* p, float v) { *p = (T)v; }
template <>
__device__ __forceinline__ void st_f32<__hip_bfloat16>(__hip_bfloat16* p, float v) {
    *p = __float2bfloat16(v);
}

// block-wide sum of two values over blockDim.x threads (<=1024), LDS based.
// Returns (s1, s2) to every thread.  `scratch` must hold 2*blockDim.x/WAVE floats.
__device__ __forceinline__ void block_reduce2(float& v1, float& v2, float* scratch) {
    // wave reduce first
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        v1 += __shfl_down(v1, off, WAVE);
        v2 += __shfl_down(v2, off, WAVE);
    }
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    if ((threadIdx.x & (WAVE - 1)) == 0) {
        scratch[2 * wid] = v1;
        scratch[2 * wid + 1] = v2;
    }
    __syncthreads();
    if (threadIdx.x < WAVE) {
        float a = (threadIdx.x < nw) ? scratch[2 * threadIdx.x] : 0.f;
        float b = (threadIdx.x < nw) ? scratch[2 * threadIdx.x + 1] : 0.f;
        for (int off = WAVE / 2; off > 0; off >>= 1) {
            a += __shfl_down(a, off, WAVE);
            b += __shfl_down(b, off, WAVE);
        }
        if (threadIdx.x == 0) { scratch[0] = a; scratch[1] = b; }
    }
    __syncthreads();
    v1 = scratch[0];
    v2 = scratch[1];
    __syncthreads();
}
