// Copyright (c) Flashy-AMD authors.
// Shared helpers for CDNA4 (gfx950) kernels.  Pure HIP — no torch headers.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define WAVE_SIZE 64

using short8 = __attribute__((ext_vector_type(8))) short;
using floatx4 = __attribute__((ext_vector_type(4))) float;

// Grid sizing for memory-bound elementwise kernels (guideline 11: cap the
// grid near 256 CU x 8 blocks and grid-stride the rest).
static inline int ew_grid(int64_t n_items, int block, int per_thread) {
    int64_t blocks = (n_items + (int64_t)block * per_thread - 1) /
                     ((int64_t)block * per_thread);
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

// bf16 <-> f32 bit helpers (torch bf16 tensors expose raw uint16 payloads)
__device__ __forceinline__ float bf16_to_f32(uint16_t u) {
    union { uint32_t u32; float f; } v;
    v.u32 = ((uint32_t)u) << 16;
    return v.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
    union { uint32_t u32; float f32; } v;
    v.f32 = f;
    // round-to-nearest-even
    uint32_t lsb = (v.u32 >> 16) & 1u;
    v.u32 += 0x7fffu + lsb;
    return (uint16_t)(v.u32 >> 16);
}

// wave-wide reductions over all 64 lanes
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, WAVE_SIZE);
    return v;  // valid in lane 0
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v = fmaxf(v, __shfl_down(v, off, WAVE_SIZE));
    return v;  // valid in lane 0
}

// broadcast lane 0 to the wave
__device__ __forceinline__ float wave_bcast(float v) {
    return __shfl(v, 0, WAVE_SIZE);
}

#define HIP_CHECK_LAST()                                                     \
    do {                                                                     \
        hipError_t err_ = hipGetLastError();                                 \
        if (err_ != hipSuccess) throw std::runtime_error(hipGetErrorString(err_)); \
    } while (0)
