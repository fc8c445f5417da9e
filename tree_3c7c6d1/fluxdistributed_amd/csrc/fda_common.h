// Common device helpers for fluxdistributed_amd gfx950 kernels.
// CDNA4: wavefront = 64 lanes; block sizes are multiples of 64.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdint>

namespace fda {

constexpr int WAVE = 64;

// ---- dtype plumbing --------------------------------------------------------
// bf16 <-> f32: load via bit-shift (exact), store via RNE intrinsic.
__device__ __forceinline__ float to_f32(float v) { return v; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }
__device__ __forceinline__ float bf16bits_to_f32(unsigned short u) {
    union { unsigned int i; float f; } c;
    c.i = (unsigned int)u << 16;
    return c.f;
}
__device__ __forceinline__ unsigned short f32_to_bf16bits(float f) {
    // round-to-nearest-even
    union { float f; unsigned int i; } c;
    c.f = f;
    unsigned int x = c.i;
    unsigned int lsb = (x >> 16) & 1u;
    x += 0x7fffu + lsb;
    return (unsigned short)(x >> 16);
}

template <typename T> struct VecWidth;
template <> struct VecWidth<float> { static constexpr int value = 4; };           // 16 B
template <> struct VecWidth<unsigned short> { static constexpr int value = 8; };  // 16 B (bf16 bits)

template <typename T>
__device__ __forceinline__ float load_f32(const T* p) {
    if constexpr (sizeof(T) == 2) return bf16bits_to_f32(*(const unsigned short*)p);
    else return *(const float*)p;
}
template <typename T>
__device__ __forceinline__ void store_f32(T* p, float v) {
    if constexpr (sizeof(T) == 2) *(unsigned short*)p = f32_to_bf16bits(v);
    else *(float*)p = v;
}

// ---- wave / block reductions ----------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
    return v;  // valid in lane 0 of the wave
}
__device__ __forceinline__ float wave_reduce_max(float v) {
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, WAVE));
    return v;
}

// Block-wide reduce for blockDim.x <= 1024 (multiple of 64). `scratch` needs
// blockDim.x/64 floats. Result broadcast to all threads.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    v = wave_reduce_sum(v);
    if (lane == 0) scratch[wid] = v;
    __syncthreads();
    float r = (lane < nw) ? scratch[lane] : 0.f;
    r = wave_reduce_sum(r);
    r = __shfl(r, 0, WAVE);
    __syncthreads();
    if (threadIdx.x == 0) scratch[0] = r;
    __syncthreads();
    return scratch[0];
}
__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    const int nw = blockDim.x / WAVE;
    v = wave_reduce_max(v);
    if (lane == 0) scratch[wid] = v;
    __syncthreads();
    float r = (lane < nw) ? scratch[lane] : -INFINITY;
    r = wave_reduce_max(r);
    r = __shfl(r, 0, WAVE);
    __syncthreads();
    if (threadIdx.x == 0) scratch[0] = r;
    __syncthreads();
    return scratch[0];
}

}  // namespace fda
