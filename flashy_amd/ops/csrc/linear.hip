// Copyright (c) Flashy-AMD authors.
// Small fully-connected (Linear) kernels for gfx950: fp32 x [B,I] @ w [O,I]^T
// + b -> y [B,O], plus the three backward products.
//
// The framework's fc layers are tiny (ResNet head 64x512 @ 10 classes,
// the basic example's Linear(32,1), the dummy teacher-student MLPs —
// SURVEY.md §2.10 "Linear fwd/bwd (tiny, plumbing path)"): the work is
// launch-bound, not FLOP-bound, so each product is ONE simple float4
// grid-strided kernel (vs torch's addmm + 2 mm + sum chain) and all four
// land in the HIP-graph-captured step like every other native op.
// Weight grads ACCUMULATE (+=) into the provided fp32 buffer — the flat
// optimizer's param.grad view — matching autograd semantics.

#include "common.h"

// y[b,o] = b[o] + sum_i x[b,i] * w[o,i]
__global__ void __launch_bounds__(256)
k_linear_fwd(const float* __restrict__ x, const float* __restrict__ w,
             const float* __restrict__ bias, float* __restrict__ y,
             int64_t B, int64_t I, int64_t O) {
    const int64_t total = B * O;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t i4 = I & ~3LL;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int64_t b = idx / O, o = idx % O;
        const float* xr = x + b * I;
        const float* wr = w + o * I;
        float acc = bias != nullptr ? bias[o] : 0.f;
        for (int64_t i = 0; i < i4; i += 4) {
            const float4 xv = *reinterpret_cast<const float4*>(xr + i);
            const float4 wv = *reinterpret_cast<const float4*>(wr + i);
            acc = fmaf(xv.x, wv.x, acc);
            acc = fmaf(xv.y, wv.y, acc);
            acc = fmaf(xv.z, wv.z, acc);
            acc = fmaf(xv.w, wv.w, acc);
        }
        for (int64_t i = i4; i < I; ++i) acc = fmaf(xr[i], wr[i], acc);
        y[idx] = acc;
    }
}

// dx[b,i] = sum_o dy[b,o] * w[o,i]
__global__ void __launch_bounds__(256)
k_linear_dx(const float* __restrict__ dy, const float* __restrict__ w,
            float* __restrict__ dx, int64_t B, int64_t I, int64_t O) {
    const int64_t total = B * I;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int64_t b = idx / I, i = idx % I;
        float acc = 0.f;
        for (int64_t o = 0; o < O; ++o)
            acc = fmaf(dy[b * O + o], w[o * I + i], acc);
        dx[idx] = acc;
    }
}

// dw[o,i] += sum_b dy[b,o] * x[b,i];   db[o] += sum_b dy[b,o]
// grid.y splits the batch: tiny O*I (e.g. the O=1 discriminator head)
// otherwise leaves most of the chip idle; splits accumulate atomically.
__global__ void __launch_bounds__(256)
k_linear_dw(const float* __restrict__ x, const float* __restrict__ dy,
            float* __restrict__ dw, float* __restrict__ db,
            int64_t B, int64_t I, int64_t O, int b_per_split) {
    const int64_t total = O * I;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t b0 = (int64_t)blockIdx.y * b_per_split;
    const int64_t b1 = min(b0 + (int64_t)b_per_split, B);
    const bool split = gridDim.y > 1;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int64_t o = idx / I, i = idx % I;
        float acc = 0.f;
        for (int64_t b = b0; b < b1; ++b)
            acc = fmaf(dy[b * O + o], x[b * I + i], acc);
        if (split) atomicAdd(&dw[idx], acc);
        else dw[idx] += acc;
        if (db != nullptr && i == 0) {
            float bs = 0.f;
            for (int64_t b = b0; b < b1; ++b) bs += dy[b * O + o];
            if (split) atomicAdd(&db[o], bs);
            else db[o] += bs;
        }
    }
}

// Wide-I, tiny-O fc forward (the DCGAN discriminator head: [B,8192] @ [1,
// 8192]^T): one 256-thread block per (b,o) dot product, float4 strided +
// LDS tree reduce.  The per-element kernel above would put B*O=64 threads
// on the whole problem; rocBLAS tiles MT16x256 against N=1 and costs
// ~90 us (profiles/r02k).
__global__ void __launch_bounds__(256)
k_linear_fwd_rows(const float* __restrict__ x, const float* __restrict__ w,
                  const float* __restrict__ bias, float* __restrict__ y,
                  int64_t B, int64_t I, int64_t O) {
    const int64_t b = blockIdx.x / O, o = blockIdx.x % O;
    const float* xr = x + b * I;
    const float* wr = w + o * I;
    const int64_t i4 = I & ~3LL;
    float acc = 0.f;
    for (int64_t i = (int64_t)threadIdx.x * 4; i < i4;
         i += (int64_t)blockDim.x * 4) {
        const float4 xv = *reinterpret_cast<const float4*>(xr + i);
        const float4 wv = *reinterpret_cast<const float4*>(wr + i);
        acc = fmaf(xv.x, wv.x, acc);
        acc = fmaf(xv.y, wv.y, acc);
        acc = fmaf(xv.z, wv.z, acc);
        acc = fmaf(xv.w, wv.w, acc);
    }
    if (threadIdx.x == 0)
        for (int64_t i = i4; i < I; ++i) acc = fmaf(xr[i], wr[i], acc);
    __shared__ float red[256];
    red[threadIdx.x] = acc;
    __syncthreads();
#pragma unroll
    for (int off = 128; off > 0; off >>= 1) {
        if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
        __syncthreads();
    }
    if (threadIdx.x == 0)
        y[b * O + o] = red[0] + (bias != nullptr ? bias[o] : 0.f);
}

extern "C" void launch_linear_fwd(const void* x, const void* w, const void* b,
                                  void* y, int64_t B, int64_t I, int64_t O,
                                  hipStream_t stream) {
    if (O <= 8 && I >= 1024 && B * O <= 65536) {
        k_linear_fwd_rows<<<(unsigned)(B * O), 256, 0, stream>>>(
            (const float*)x, (const float*)w, (const float*)b, (float*)y,
            B, I, O);
        return;
    }
    k_linear_fwd<<<ew_grid(B * O, 256, 1), 256, 0, stream>>>(
        (const float*)x, (const float*)w, (const float*)b, (float*)y, B, I, O);
}

extern "C" void launch_linear_dx(const void* dy, const void* w, void* dx,
                                 int64_t B, int64_t I, int64_t O,
                                 hipStream_t stream) {
    k_linear_dx<<<ew_grid(B * I, 256, 1), 256, 0, stream>>>(
        (const float*)dy, (const float*)w, (float*)dx, B, I, O);
}

extern "C" void launch_linear_dw(const void* x, const void* dy, void* dw,
                                 void* db, int64_t B, int64_t I, int64_t O,
                                 hipStream_t stream) {
    const int gx = ew_grid(O * I, 256, 1);
    int zn = 1;
    if (gx < 160 && B >= 8) {
        zn = (int)(160 / gx > 8 ? 8 : 160 / gx);
        if (zn > (int)B / 4) zn = (int)B / 4 > 0 ? (int)B / 4 : 1;
        if (zn < 1) zn = 1;
    }
    const int bps = (int)((B + zn - 1) / zn);
    dim3 grid((unsigned)gx, (unsigned)zn);
    k_linear_dw<<<grid, 256, 0, stream>>>(
        (const float*)x, (const float*)dy, (float*)dw, (float*)db, B, I, O,
        bps);
}
