// Copyright (c) Flashy-AMD authors.
// NHWC BatchNorm (training) for gfx950, operating on [M, C] with
// M = N*H*W, bf16 activations, fp32 stats/params.
//
// forward:  stats (per-channel sum/sumsq partials; short8 loads, one
//                  partial slot per block — no atomics, no pre-zeroing)
//        -> finalize (combine partials, mean/invstd/scale/shift, running)
//        -> apply (y = relu(scale*x + shift [+ res]), float4 param loads)
// backward: reduce (dz = dy * relu-mask; partial sums; dz written out —
//                   it IS the residual gradient)
//        -> grads (combine -> bsums; dgamma/dbeta += into flat fp32 grads)
//        -> apply (dx = scale*(dz - (sum_dz + xhat*sum_dzxhat)/M))
//
// partials layout [2][C][msplit] so the combine kernels read each channel's
// msplit partials CONTIGUOUSLY.
//
// Replaces the BatchNorm + ReLU + residual-add chains of the reference's
// ResNet workload with NHWC-native fused kernels.

#include "common.h"

// ---------------------------------------------------------------------------
// fwd 1: stats.  Block = 256 threads = 8 channel-octets x 32 m-lanes over a
// 64-channel group; short8 loads (16 B/lane).  Grid (C/64, msplit).
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
k_bn_stats(const uint16_t* __restrict__ x, float* __restrict__ partials,
           int64_t M, int C, int msplit, int m_per_block) {
    const int c8 = (threadIdx.x & 7) * 8;           // channel octet in group
    const int mlane = threadIdx.x >> 3;             // 0..31
    const int cbase = blockIdx.x * 64;
    const int64_t m0 = (int64_t)blockIdx.y * m_per_block;
    const int64_t m1 = min(m0 + (int64_t)m_per_block, M);
    float s[8] = {}, s2[8] = {};
    for (int64_t m = m0 + mlane; m < m1; m += 32) {
        const short8 v8 = *reinterpret_cast<const short8*>(
            x + m * C + cbase + c8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float v = bf16_to_f32(((const uint16_t*)&v8)[j]);
            s[j] += v;
            s2[j] = fmaf(v, v, s2[j]);
        }
    }
    // fold 32 m-lanes: LDS [2][32][64]
    __shared__ float red[2][32][64];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        red[0][mlane][c8 + j] = s[j];
        red[1][mlane][c8 + j] = s2[j];
    }
    __syncthreads();
    // 256 threads: tid<64 handles sum, tid in [64,128) handles sumsq
    const int c = threadIdx.x & 63;
    const int which = threadIdx.x >> 6;
    if (which < 2) {
        float acc = 0.f;
#pragma unroll 8
        for (int i = 0; i < 32; ++i) acc += red[which][i][c];
        partials[((int64_t)which * C + cbase + c) * msplit + blockIdx.y] = acc;
    }
}

// ---------------------------------------------------------------------------
// combine helper: [2][C][msplit] -> (s, s2) per channel.  Block = 256
// threads = 64 channels x 4 split-lanes; contiguous per-channel reads.
// ---------------------------------------------------------------------------

// Block = 8 channels x 32 split-lanes (grid C/8): at C=64 that is 8 blocks
// instead of 1, and each thread issues only ~msplit/16 float4 loads, so the
// single-digit-block latency wall of a C/64 mapping disappears.  Threads
// tid<8 leave with the channel's (s, s2); they are also the writers.
__device__ __forceinline__ void combine_partials8(
        const float* __restrict__ partials, int msplit, int C,
        float* s_out, float* s2_out) {
    const int ch = threadIdx.x & 7;               // channel within block
    const int slane = threadIdx.x >> 3;           // 0..31
    const int c = blockIdx.x * 8 + ch;
    const float* row0 = partials + (int64_t)c * msplit;
    const float* row1 = partials + ((int64_t)C + c) * msplit;
    float s = 0.f, s2 = 0.f;
    if (msplit & 3) {
        // row base c*msplit is 16B-aligned only when msplit % 4 == 0 (the
        // fused conv-epilogue path passes msplit = conv grid.x, which is
        // not forced to a multiple of 4) -> scalar loads for odd msplit.
        for (int i = slane; i < msplit; i += 32) {
            s += row0[i];
            s2 += row1[i];
        }
    } else {
        for (int i = slane * 4; i + 4 <= msplit; i += 128) {
            const float4 a = *reinterpret_cast<const float4*>(row0 + i);
            const float4 b = *reinterpret_cast<const float4*>(row1 + i);
            s += a.x + a.y + a.z + a.w;
            s2 += b.x + b.y + b.z + b.w;
        }
    }
    __shared__ float red[2][32][8];
    red[0][slane][ch] = s;
    red[1][slane][ch] = s2;
    __syncthreads();
    if (threadIdx.x < 16) {
        const int w = threadIdx.x >> 3;
        const int cc = threadIdx.x & 7;
        float acc = 0.f;
#pragma unroll 8
        for (int i = 0; i < 32; ++i) acc += red[w][i][cc];
        red[w][0][cc] = acc;
    }
    __syncthreads();
    *s_out = red[0][0][ch];
    *s2_out = red[1][0][ch];
}

// work[0..C) = mean, [C..2C) = invstd, [2C..3C) = scale, [3C..4C) = shift
__global__ void __launch_bounds__(256)
k_bn_finalize(const float* __restrict__ partials, int msplit,
              const float* __restrict__ gamma,
              const float* __restrict__ beta,
              float* __restrict__ running_mean,
              float* __restrict__ running_var,
              float* __restrict__ work, int64_t M, int C,
              float eps, float momentum, int update_running) {
    float s, s2;
    combine_partials8(partials, msplit, C, &s, &s2);
    if (threadIdx.x >= 8) return;
    const int c = blockIdx.x * 8 + threadIdx.x;
    const float mean = s / (float)M;
    float var = s2 / (float)M - mean * mean;
    var = fmaxf(var, 0.f);
    const float invstd = rsqrtf(var + eps);
    const float scale = gamma[c] * invstd;
    work[c] = mean;
    work[C + c] = invstd;
    work[2 * C + c] = scale;
    work[3 * C + c] = beta[c] - mean * scale;
    if (update_running) {
        running_mean[c] += momentum * (mean - running_mean[c]);
        const float unbiased = M > 1 ? var * (float)M / (float)(M - 1) : var;
        running_var[c] += momentum * (unbiased - running_var[c]);
    }
}

// ---------------------------------------------------------------------------
// fwd 3: y = [relu](scale*x + shift [+ res]).  short8 data, float4 params.
// ---------------------------------------------------------------------------

template <bool RELU, bool RES>
__global__ void __launch_bounds__(256)
k_bn_apply(const uint16_t* __restrict__ x, const uint16_t* __restrict__ res,
           uint16_t* __restrict__ y, const float* __restrict__ work,
           int64_t M, int C, float slope) {
    const int64_t total8 = M * C / 8;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const float4* scale4 = reinterpret_cast<const float4*>(work + 2 * (int64_t)C);
    const float4* shift4 = reinterpret_cast<const float4*>(work + 3 * (int64_t)C);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total8;
         i += stride) {
        const int co = (int)((i * 8) % C) / 4;
        const float4 sc[2] = {scale4[co], scale4[co + 1]};
        const float4 sh[2] = {shift4[co], shift4[co + 1]};
        short8 xv = *reinterpret_cast<const short8*>(x + i * 8);
        short8 rv = {};
        if (RES) rv = *reinterpret_cast<const short8*>(res + i * 8);
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float v = fmaf(bf16_to_f32(((const uint16_t*)&xv)[j]),
                           ((const float*)sc)[j], ((const float*)sh)[j]);
            if (RES) v += bf16_to_f32(((const uint16_t*)&rv)[j]);
            if (RELU) v = v > 0.f ? v : slope * v;  // slope 0 = plain ReLU
            ((uint16_t*)&out)[j] = f32_to_bf16(v);
        }
        *reinterpret_cast<short8*>(y + i * 8) = out;
    }
}

// Channel-resident variant for C % 64 == 0 (every trunk BN): thread owns a
// fixed 8-channel group, coefficients load ONCE into registers, the loop is
// pure 16B streaming over m — no per-iteration int64 div/mod (the generic
// kernel's `(i*8) % C` is an emulated 64-bit divide and caps it ~3 TB/s).
template <bool RELU, bool RES>
__global__ void __launch_bounds__(256)
k_bn_apply_c64(const uint16_t* __restrict__ x, const uint16_t* __restrict__ res,
               uint16_t* __restrict__ y, const float* __restrict__ work,
               int64_t M, int C, float slope, int m_per_block) {
    const int c = blockIdx.x * 64 + (threadIdx.x & 7) * 8;
    const int mlane = threadIdx.x >> 3;
    const int64_t m0 = (int64_t)blockIdx.y * m_per_block;
    const int64_t m1 = min(m0 + (int64_t)m_per_block, M);
    float sc[8], sh[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        sc[j] = work[2 * (int64_t)C + c + j];
        sh[j] = work[3 * (int64_t)C + c + j];
    }
    for (int64_t m = m0 + mlane; m < m1; m += 32) {
        const int64_t off = m * C + c;
        short8 xv = *reinterpret_cast<const short8*>(x + off);
        short8 rv = {};
        if (RES) rv = *reinterpret_cast<const short8*>(res + off);
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float v = fmaf(bf16_to_f32(((const uint16_t*)&xv)[j]), sc[j], sh[j]);
            if (RES) v += bf16_to_f32(((const uint16_t*)&rv)[j]);
            if (RELU) v = v > 0.f ? v : slope * v;
            ((uint16_t*)&out)[j] = f32_to_bf16(v);
        }
        *reinterpret_cast<short8*>(y + off) = out;
    }
}

__global__ void __launch_bounds__(256)
k_bn_bwd_apply_c64(const uint16_t* __restrict__ dz,
                   const uint16_t* __restrict__ x,
                   const float* __restrict__ work,
                   const float* __restrict__ bsums,
                   uint16_t* __restrict__ dx, int64_t M, int C,
                   int m_per_block) {
    const int c = blockIdx.x * 64 + (threadIdx.x & 7) * 8;
    const int mlane = threadIdx.x >> 3;
    const int64_t m0 = (int64_t)blockIdx.y * m_per_block;
    const int64_t m1 = min(m0 + (int64_t)m_per_block, M);
    const float invM = 1.f / (float)M;
    float mn[8], is[8], sc[8], b0[8], b1[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        mn[j] = work[c + j];
        is[j] = work[C + c + j];
        sc[j] = work[2 * (int64_t)C + c + j];
        b0[j] = bsums[c + j] * invM;
        b1[j] = bsums[C + c + j] * invM;
    }
    for (int64_t m = m0 + mlane; m < m1; m += 32) {
        const int64_t off = m * C + c;
        short8 gz = *reinterpret_cast<const short8*>(dz + off);
        short8 xv = *reinterpret_cast<const short8*>(x + off);
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float xhat = (bf16_to_f32(((const uint16_t*)&xv)[j]) - mn[j])
                               * is[j];
            const float g = bf16_to_f32(((const uint16_t*)&gz)[j]);
            ((uint16_t*)&out)[j] =
                f32_to_bf16(sc[j] * (g - b0[j] - xhat * b1[j]));
        }
        *reinterpret_cast<short8*>(dx + off) = out;
    }
}

// m-blocks for the c64 apply kernels: ~1024 blocks saturate the chip
static inline int apply_mpb(int64_t M, int C) {
    const int cblocks = C / 64;
    int target = 1024 / cblocks;
    if (target < 1) target = 1;
    int64_t mblocks = (M + 31) / 32;
    if (mblocks > target) mblocks = target;
    return (int)((M + mblocks - 1) / mblocks);
}

// ---------------------------------------------------------------------------
// bwd 1: dz = dy * (y > 0) [if relu]; per-channel partial sums of dz and
// dz*xhat.  Same geometry as stats.
// ---------------------------------------------------------------------------

template <bool RELU>
__global__ void __launch_bounds__(256)
k_bn_bwd_reduce(const uint16_t* __restrict__ dy, const uint16_t* __restrict__ y,
                const uint16_t* __restrict__ x, const float* __restrict__ work,
                uint16_t* __restrict__ dz_out, float* __restrict__ partials,
                int64_t M, int C, int msplit, int m_per_block, float slope) {
    const int c8 = (threadIdx.x & 7) * 8;
    const int mlane = threadIdx.x >> 3;
    const int cbase = blockIdx.x * 64;
    const int64_t m0 = (int64_t)blockIdx.y * m_per_block;
    const int64_t m1 = min(m0 + (int64_t)m_per_block, M);
    float mean[8], invstd[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        mean[j] = work[cbase + c8 + j];
        invstd[j] = work[C + cbase + c8 + j];
    }
    float s[8] = {}, sx[8] = {};
    for (int64_t m = m0 + mlane; m < m1; m += 32) {
        const int64_t off = m * C + cbase + c8;
        const short8 g8 = *reinterpret_cast<const short8*>(dy + off);
        const short8 x8 = *reinterpret_cast<const short8*>(x + off);
        short8 y8 = {};
        if (RELU) y8 = *reinterpret_cast<const short8*>(y + off);
        short8 dz8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float g = bf16_to_f32(((const uint16_t*)&g8)[j]);
            if (RELU && bf16_to_f32(((const uint16_t*)&y8)[j]) <= 0.f)
                g *= slope;
            ((uint16_t*)&dz8)[j] = f32_to_bf16(g);
            s[j] += g;
            const float xh = (bf16_to_f32(((const uint16_t*)&x8)[j]) - mean[j]) * invstd[j];
            sx[j] = fmaf(g, xh, sx[j]);
        }
        *reinterpret_cast<short8*>(dz_out + off) = dz8;
    }
    __shared__ float red[2][32][64];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        red[0][mlane][c8 + j] = s[j];
        red[1][mlane][c8 + j] = sx[j];
    }
    __syncthreads();
    const int c = threadIdx.x & 63;
    const int which = threadIdx.x >> 6;
    if (which < 2) {
        float acc = 0.f;
#pragma unroll 8
        for (int i = 0; i < 32; ++i) acc += red[which][i][c];
        partials[((int64_t)which * C + cbase + c) * msplit + blockIdx.y] = acc;
    }
}

// combine partials -> bsums[2C]; dgamma/dbeta += (flat fp32 grads)
__global__ void __launch_bounds__(256)
k_bn_bwd_grads(const float* __restrict__ partials, int msplit,
               float* __restrict__ bsums,
               float* __restrict__ dgamma,
               float* __restrict__ dbeta, int C) {
    float s, sx;
    combine_partials8(partials, msplit, C, &s, &sx);
    if (threadIdx.x >= 8) return;
    const int c = blockIdx.x * 8 + threadIdx.x;
    bsums[c] = s;
    bsums[C + c] = sx;
    dbeta[c] += s;
    dgamma[c] += sx;
}

// ---------------------------------------------------------------------------
// bwd 3: dx = scale * (dz - (sum_dz + xhat * sum_dz_xhat) / M)
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
k_bn_bwd_apply(const uint16_t* __restrict__ dz, const uint16_t* __restrict__ x,
               const float* __restrict__ work, const float* __restrict__ bsums,
               uint16_t* __restrict__ dx, int64_t M, int C) {
    const int64_t total8 = M * C / 8;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const float invM = 1.f / (float)M;
    const float4* mean4 = reinterpret_cast<const float4*>(work);
    const float4* invstd4 = reinterpret_cast<const float4*>(work + C);
    const float4* scale4 = reinterpret_cast<const float4*>(work + 2 * (int64_t)C);
    const float4* b04 = reinterpret_cast<const float4*>(bsums);
    const float4* b14 = reinterpret_cast<const float4*>(bsums + C);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total8;
         i += stride) {
        const int co = (int)((i * 8) % C) / 4;
        const float4 mn[2] = {mean4[co], mean4[co + 1]};
        const float4 is[2] = {invstd4[co], invstd4[co + 1]};
        const float4 sc[2] = {scale4[co], scale4[co + 1]};
        const float4 b0[2] = {b04[co], b04[co + 1]};
        const float4 b1[2] = {b14[co], b14[co + 1]};
        short8 gz = *reinterpret_cast<const short8*>(dz + i * 8);
        short8 xv = *reinterpret_cast<const short8*>(x + i * 8);
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float xhat = (bf16_to_f32(((const uint16_t*)&xv)[j]) -
                                ((const float*)mn)[j]) * ((const float*)is)[j];
            const float g = bf16_to_f32(((const uint16_t*)&gz)[j]);
            const float v = ((const float*)sc)[j] *
                (g - (((const float*)b0)[j] + xhat * ((const float*)b1)[j]) * invM);
            ((uint16_t*)&out)[j] = f32_to_bf16(v);
        }
        *reinterpret_cast<short8*>(dx + i * 8) = out;
    }
}

// ---------------------------------------------------------------------------
// launchers (msplit chosen by the Python wrapper, shared by both phases)
// ---------------------------------------------------------------------------

extern "C" void launch_bn_stats(const void* x, void* partials, int64_t M, int C,
                                int msplit, hipStream_t stream) {
    const int mpb = (int)((M + msplit - 1) / msplit);
    dim3 grid((unsigned)(C / 64), (unsigned)msplit);
    k_bn_stats<<<grid, 256, 0, stream>>>((const uint16_t*)x, (float*)partials,
                                         M, C, msplit, mpb);
}

extern "C" void launch_bn_finalize(const void* partials, int msplit,
                                   const void* gamma, const void* beta,
                                   void* running_mean, void* running_var,
                                   void* work, int64_t M, int C, float eps,
                                   float momentum, int update_running,
                                   hipStream_t stream) {
    k_bn_finalize<<<C / 8, 256, 0, stream>>>(
        (const float*)partials, msplit, (const float*)gamma,
        (const float*)beta, (float*)running_mean, (float*)running_var,
        (float*)work, M, C, eps, momentum, update_running);
}

extern "C" void launch_bn_apply(const void* x, const void* res, void* y,
                                const void* work, int64_t M, int C, int relu,
                                float slope, hipStream_t stream) {
    if (C % 64 == 0) {
        const int mpb = apply_mpb(M, C);
        dim3 g((unsigned)(C / 64), (unsigned)((M + mpb - 1) / mpb));
        if (relu && res)
            k_bn_apply_c64<true, true><<<g, 256, 0, stream>>>(
                (const uint16_t*)x, (const uint16_t*)res, (uint16_t*)y,
                (const float*)work, M, C, slope, mpb);
        else if (relu)
            k_bn_apply_c64<true, false><<<g, 256, 0, stream>>>(
                (const uint16_t*)x, nullptr, (uint16_t*)y,
                (const float*)work, M, C, slope, mpb);
        else if (res)
            k_bn_apply_c64<false, true><<<g, 256, 0, stream>>>(
                (const uint16_t*)x, (const uint16_t*)res, (uint16_t*)y,
                (const float*)work, M, C, slope, mpb);
        else
            k_bn_apply_c64<false, false><<<g, 256, 0, stream>>>(
                (const uint16_t*)x, nullptr, (uint16_t*)y,
                (const float*)work, M, C, slope, mpb);
        return;
    }
    const int grid = ew_grid(M * C / 8, 256, 4);
    if (relu && res)
        k_bn_apply<true, true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)res, (uint16_t*)y,
            (const float*)work, M, C, slope);
    else if (relu)
        k_bn_apply<true, false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, nullptr, (uint16_t*)y, (const float*)work, M, C, slope);
    else if (res)
        k_bn_apply<false, true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)res, (uint16_t*)y,
            (const float*)work, M, C, slope);
    else
        k_bn_apply<false, false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, nullptr, (uint16_t*)y, (const float*)work, M, C, slope);
}

extern "C" void launch_bn_bwd_reduce(const void* dy, const void* y,
                                     const void* x, const void* work,
                                     void* dz_out, void* partials, int64_t M,
                                     int C, int msplit, int relu, float slope,
                                     hipStream_t stream) {
    const int mpb = (int)((M + msplit - 1) / msplit);
    dim3 grid((unsigned)(C / 64), (unsigned)msplit);
    if (relu)
        k_bn_bwd_reduce<true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dy, (const uint16_t*)y, (const uint16_t*)x,
            (const float*)work, (uint16_t*)dz_out, (float*)partials, M, C,
            msplit, mpb, slope);
    else
        k_bn_bwd_reduce<false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dy, (const uint16_t*)y, (const uint16_t*)x,
            (const float*)work, (uint16_t*)dz_out, (float*)partials, M, C,
            msplit, mpb, slope);
}

extern "C" void launch_bn_bwd_grads(const void* partials, int msplit,
                                    void* bsums, void* dgamma, void* dbeta,
                                    int C, hipStream_t stream) {
    k_bn_bwd_grads<<<C / 8, 256, 0, stream>>>(
        (const float*)partials, msplit, (float*)bsums, (float*)dgamma,
        (float*)dbeta, C);
}

extern "C" void launch_bn_bwd_apply(const void* dz, const void* x,
                                    const void* work, const void* bsums,
                                    void* dx, int64_t M, int C,
                                    hipStream_t stream) {
    if (C % 64 == 0) {
        const int mpb = apply_mpb(M, C);
        dim3 g((unsigned)(C / 64), (unsigned)((M + mpb - 1) / mpb));
        k_bn_bwd_apply_c64<<<g, 256, 0, stream>>>(
            (const uint16_t*)dz, (const uint16_t*)x, (const float*)work,
            (const float*)bsums, (uint16_t*)dx, M, C, mpb);
        return;
    }
    const int grid = ew_grid(M * C / 8, 256, 4);
    k_bn_bwd_apply<<<grid, 256, 0, stream>>>(
        (const uint16_t*)dz, (const uint16_t*)x, (const float*)work,
        (const float*)bsums, (uint16_t*)dx, M, C);
}
