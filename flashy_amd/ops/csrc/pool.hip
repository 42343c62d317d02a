// Copyright (c) Flashy-AMD authors.
// NHWC max-pooling for gfx950 (the ImageNet ResNet stem: 3x3 stride 2).
//
// forward: per (output pixel, channel-octet) thread, short8 loads over the
// window; stores bf16 maxima plus a uint8 argmax window-position per
// (output, channel) for the backward.
// backward: gather-based (deterministic, no atomics): each INPUT pixel
// checks the <= (R/stride+1)^2 windows that cover it and sums the dy of
// windows whose stored argmax points at it.

#include "conv_common.h"

__global__ void __launch_bounds__(256)
k_maxpool_fwd(const uint16_t* __restrict__ x, uint16_t* __restrict__ y,
              uint8_t* __restrict__ argmax, ConvDims d) {
    const int64_t total = (int64_t)d.N * d.Ho * d.Wo * (d.C / 8);
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int c8 = (int)(idx % (d.C / 8)) * 8;
        const int64_t m = idx / (d.C / 8);
        const int wo = (int)(m % d.Wo);
        const int ho = (int)((m / d.Wo) % d.Ho);
        const int64_t n = m / ((int64_t)d.Ho * d.Wo);
        float best[8];
        int arg[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) { best[j] = -INFINITY; arg[j] = 0; }
        for (int r = 0; r < d.R; ++r) {
            const int hi = ho * d.stride + r - d.pad;
            if (hi < 0 || hi >= d.H) continue;
            for (int s = 0; s < d.S; ++s) {
                const int wi = wo * d.stride + s - d.pad;
                if (wi < 0 || wi >= d.W) continue;
                const short8 v = *reinterpret_cast<const short8*>(
                    x + ((n * d.H + hi) * d.W + wi) * (int64_t)d.C + c8);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    const float f = bf16_to_f32(((const uint16_t*)&v)[j]);
                    if (f > best[j]) { best[j] = f; arg[j] = r * d.S + s; }
                }
            }
        }
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            ((uint16_t*)&out)[j] = f32_to_bf16(best[j]);
            argmax[m * d.C + c8 + j] = (uint8_t)arg[j];
        }
        *reinterpret_cast<short8*>(y + m * d.C + c8) = out;
    }
}

__global__ void __launch_bounds__(256)
k_maxpool_bwd(const uint16_t* __restrict__ dy, const uint8_t* __restrict__ argmax,
              uint16_t* __restrict__ dx, ConvDims d) {
    const int64_t total = (int64_t)d.N * d.H * d.W * (d.C / 8);
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int c8 = (int)(idx % (d.C / 8)) * 8;
        const int64_t m = idx / (d.C / 8);
        const int wi = (int)(m % d.W);
        const int hi = (int)((m / d.W) % d.H);
        const int64_t n = m / ((int64_t)d.H * d.W);
        float acc[8] = {};
        // windows (ho, wo) covering (hi, wi): ho*stride + r - pad == hi
        for (int r = 0; r < d.R; ++r) {
            const int hnum = hi + d.pad - r;
            if (hnum < 0 || hnum % d.stride) continue;
            const int ho = hnum / d.stride;
            if (ho >= d.Ho) continue;
            for (int s = 0; s < d.S; ++s) {
                const int wnum = wi + d.pad - s;
                if (wnum < 0 || wnum % d.stride) continue;
                const int wo = wnum / d.stride;
                if (wo >= d.Wo) continue;
                const int64_t mo = (n * d.Ho + ho) * d.Wo + wo;
                const short8 g = *reinterpret_cast<const short8*>(
                    dy + mo * d.C + c8);
                const int pos = r * d.S + s;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    if (argmax[mo * d.C + c8 + j] == pos)
                        acc[j] += bf16_to_f32(((const uint16_t*)&g)[j]);
            }
        }
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) ((uint16_t*)&out)[j] = f32_to_bf16(acc[j]);
        *reinterpret_cast<short8*>(dx + m * d.C + c8) = out;
    }
}

extern "C" void launch_maxpool_fwd(const void* x, void* y, void* argmax,
                                   ConvDims d, hipStream_t stream) {
    const int64_t total = (int64_t)d.N * d.Ho * d.Wo * (d.C / 8);
    k_maxpool_fwd<<<ew_grid(total, 256, 2), 256, 0, stream>>>(
        (const uint16_t*)x, (uint16_t*)y, (uint8_t*)argmax, d);
}

extern "C" void launch_maxpool_bwd(const void* dy, const void* argmax, void* dx,
                                   ConvDims d, hipStream_t stream) {
    const int64_t total = (int64_t)d.N * d.H * d.W * (d.C / 8);
    k_maxpool_bwd<<<ew_grid(total, 256, 2), 256, 0, stream>>>(
        (const uint16_t*)dy, (const uint8_t*)argmax, (uint16_t*)dx, d);
}
