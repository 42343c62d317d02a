// Copyright (c) Flashy-AMD authors.
// NHWC BatchNorm (training) for gfx950, operating on [M, C] with
// M = N*H*W, bf16 activations, fp32 stats/params.
//
// forward:  stats (per-channel sum/sumsq partials, one slot per block —
//                  no atomics, no pre-zeroed buffers)
//        -> finalize (tiny: combine partials, mean/invstd/scale/shift,
//                     running stats)
//        -> apply (y = relu(scale*x + shift [+ res]))  [fused add+ReLU]
// backward: reduce (dz = dy * relu-mask; per-channel partial sums;
//                   dz written out — it IS the residual gradient)
//        -> grads (tiny: combine partials -> bsums; dgamma/dbeta += into
//                  flat fp32 grads)
//        -> apply (dx = scale*(dz - (sum_dz + xhat*sum_dzxhat)/M))
//
// Replaces the BatchNorm + ReLU + residual-add chains of the reference's
// ResNet workload with NHWC-native fused kernels.

#include "common.h"

// partials layout: [msplit][2][C]  (sum, then sumsq/dzxhat)

__global__ void __launch_bounds__(256)
k_bn_stats(const uint16_t* __restrict__ x, float* __restrict__ partials,
           int64_t M, int C, int m_per_block) {
    const int c = blockIdx.x * 64 + (threadIdx.x & 63);
    const int mlane = threadIdx.x >> 6;  // 0..3
    const int64_t m0 = (int64_t)blockIdx.y * m_per_block;
    const int64_t m1 = min(m0 + (int64_t)m_per_block, M);
    float s = 0.f, s2 = 0.f;
    for (int64_t m = m0 + mlane; m < m1; m += 4) {
        const float v = bf16_to_f32(x[m * C + c]);
        s += v;
        s2 = fmaf(v, v, s2);
    }
    __shared__ float red[2][4][64];
    red[0][mlane][threadIdx.x & 63] = s;
    red[1][mlane][threadIdx.x & 63] = s2;
    __syncthreads();
    if (mlane == 0) {
        s = red[0][0][threadIdx.x] + red[0][1][threadIdx.x] +
            red[0][2][threadIdx.x] + red[0][3][threadIdx.x];
        s2 = red[1][0][threadIdx.x] + red[1][1][threadIdx.x] +
             red[1][2][threadIdx.x] + red[1][3][threadIdx.x];
        float* slot = partials + (int64_t)blockIdx.y * 2 * C;
        slot[c] = s;
        slot[C + c] = s2;
    }
}

// Parallel combine of [msplit][2][C] partials: 256 threads = 64 channels x
// 4 split-lanes, many loads in flight, LDS reduce.  Grid C/64.
__device__ __forceinline__ void combine_partials(
        const float* __restrict__ partials, int msplit, int C,
        float* s_out, float* s2_out) {
    const int c = blockIdx.x * 64 + (threadIdx.x & 63);
    const int slane = threadIdx.x >> 6;  // 0..3
    float s = 0.f, s2 = 0.f;
    for (int i = slane; i < msplit; i += 4) {
        s += partials[(int64_t)i * 2 * C + c];
        s2 += partials[(int64_t)i * 2 * C + C + c];
    }
    __shared__ float red[2][4][64];
    red[0][slane][threadIdx.x & 63] = s;
    red[1][slane][threadIdx.x & 63] = s2;
    __syncthreads();
    *s_out = red[0][0][threadIdx.x & 63] + red[0][1][threadIdx.x & 63] +
             red[0][2][threadIdx.x & 63] + red[0][3][threadIdx.x & 63];
    *s2_out = red[1][0][threadIdx.x & 63] + red[1][1][threadIdx.x & 63] +
              red[1][2][threadIdx.x & 63] + red[1][3][threadIdx.x & 63];
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
    combine_partials(partials, msplit, C, &s, &s2);
    if (threadIdx.x >= 64) return;
    const int c = blockIdx.x * 64 + threadIdx.x;
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

template <bool RELU, bool RES>
__global__ void __launch_bounds__(256)
k_bn_apply(const uint16_t* __restrict__ x, const uint16_t* __restrict__ res,
           uint16_t* __restrict__ y, const float* __restrict__ work,
           int64_t M, int C) {
    const int64_t total8 = M * C / 8;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const float* scale = work + 2 * (int64_t)C;
    const float* shift = work + 3 * (int64_t)C;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total8;
         i += stride) {
        const int c0 = (int)((i * 8) % C);
        short8 xv = *reinterpret_cast<const short8*>(x + i * 8);
        short8 rv = {};
        if (RES) rv = *reinterpret_cast<const short8*>(res + i * 8);
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float v = fmaf(bf16_to_f32(((const uint16_t*)&xv)[j]),
                           scale[c0 + j], shift[c0 + j]);
            if (RES) v += bf16_to_f32(((const uint16_t*)&rv)[j]);
            if (RELU) v = fmaxf(v, 0.f);
            ((uint16_t*)&out)[j] = f32_to_bf16(v);
        }
        *reinterpret_cast<short8*>(y + i * 8) = out;
    }
}

template <bool RELU>
__global__ void __launch_bounds__(256)
k_bn_bwd_reduce(const uint16_t* __restrict__ dy, const uint16_t* __restrict__ y,
                const uint16_t* __restrict__ x, const float* __restrict__ work,
                uint16_t* __restrict__ dz_out, float* __restrict__ partials,
                int64_t M, int C, int m_per_block) {
    const int c = blockIdx.x * 64 + (threadIdx.x & 63);
    const int mlane = threadIdx.x >> 6;
    const int64_t m0 = (int64_t)blockIdx.y * m_per_block;
    const int64_t m1 = min(m0 + (int64_t)m_per_block, M);
    const float mean = work[c];
    const float invstd = work[C + c];
    float s = 0.f, sx = 0.f;
    for (int64_t m = m0 + mlane; m < m1; m += 4) {
        const int64_t i = m * C + c;
        float g = bf16_to_f32(dy[i]);
        if (RELU && bf16_to_f32(y[i]) <= 0.f) g = 0.f;
        dz_out[i] = f32_to_bf16(g);
        s += g;
        sx = fmaf(g, (bf16_to_f32(x[i]) - mean) * invstd, sx);
    }
    __shared__ float red[2][4][64];
    red[0][mlane][threadIdx.x & 63] = s;
    red[1][mlane][threadIdx.x & 63] = sx;
    __syncthreads();
    if (mlane == 0) {
        s = red[0][0][threadIdx.x] + red[0][1][threadIdx.x] +
            red[0][2][threadIdx.x] + red[0][3][threadIdx.x];
        sx = red[1][0][threadIdx.x] + red[1][1][threadIdx.x] +
             red[1][2][threadIdx.x] + red[1][3][threadIdx.x];
        float* slot = partials + (int64_t)blockIdx.y * 2 * C;
        slot[c] = s;
        slot[C + c] = sx;
    }
}

// combine partials -> bsums[2C]; dgamma/dbeta += (flat fp32 grads)
__global__ void __launch_bounds__(256)
k_bn_bwd_grads(const float* __restrict__ partials, int msplit,
               float* __restrict__ bsums,
               float* __restrict__ dgamma,
               float* __restrict__ dbeta, int C) {
    float s, sx;
    combine_partials(partials, msplit, C, &s, &sx);
    if (threadIdx.x >= 64) return;
    const int c = blockIdx.x * 64 + threadIdx.x;
    bsums[c] = s;
    bsums[C + c] = sx;
    dbeta[c] += s;
    dgamma[c] += sx;
}

__global__ void __launch_bounds__(256)
k_bn_bwd_apply(const uint16_t* __restrict__ dz, const uint16_t* __restrict__ x,
               const float* __restrict__ work, const float* __restrict__ bsums,
               uint16_t* __restrict__ dx, int64_t M, int C) {
    const int64_t total8 = M * C / 8;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const float invM = 1.f / (float)M;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total8;
         i += stride) {
        const int c0 = (int)((i * 8) % C);
        short8 gz = *reinterpret_cast<const short8*>(dz + i * 8);
        short8 xv = *reinterpret_cast<const short8*>(x + i * 8);
        short8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const int c = c0 + j;
            const float mean = work[c];
            const float invstd = work[C + c];
            const float scale = work[2 * C + c];
            const float xhat = (bf16_to_f32(((const uint16_t*)&xv)[j]) - mean) * invstd;
            const float g = bf16_to_f32(((const uint16_t*)&gz)[j]);
            const float v = scale * (g - (bsums[c] + xhat * bsums[C + c]) * invM);
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
                                         M, C, mpb);
}

extern "C" void launch_bn_finalize(const void* partials, int msplit,
                                   const void* gamma, const void* beta,
                                   void* running_mean, void* running_var,
                                   void* work, int64_t M, int C, float eps,
                                   float momentum, int update_running,
                                   hipStream_t stream) {
    k_bn_finalize<<<C / 64, 256, 0, stream>>>(
        (const float*)partials, msplit, (const float*)gamma,
        (const float*)beta, (float*)running_mean, (float*)running_var,
        (float*)work, M, C, eps, momentum, update_running);
}

extern "C" void launch_bn_apply(const void* x, const void* res, void* y,
                                const void* work, int64_t M, int C, int relu,
                                hipStream_t stream) {
    const int grid = ew_grid(M * C / 8, 256, 4);
    if (relu && res)
        k_bn_apply<true, true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)res, (uint16_t*)y,
            (const float*)work, M, C);
    else if (relu)
        k_bn_apply<true, false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, nullptr, (uint16_t*)y, (const float*)work, M, C);
    else if (res)
        k_bn_apply<false, true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)res, (uint16_t*)y,
            (const float*)work, M, C);
    else
        k_bn_apply<false, false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, nullptr, (uint16_t*)y, (const float*)work, M, C);
}

extern "C" void launch_bn_bwd_reduce(const void* dy, const void* y,
                                     const void* x, const void* work,
                                     void* dz_out, void* partials, int64_t M,
                                     int C, int msplit, int relu,
                                     hipStream_t stream) {
    const int mpb = (int)((M + msplit - 1) / msplit);
    dim3 grid((unsigned)(C / 64), (unsigned)msplit);
    if (relu)
        k_bn_bwd_reduce<true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dy, (const uint16_t*)y, (const uint16_t*)x,
            (const float*)work, (uint16_t*)dz_out, (float*)partials, M, C, mpb);
    else
        k_bn_bwd_reduce<false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dy, (const uint16_t*)y, (const uint16_t*)x,
            (const float*)work, (uint16_t*)dz_out, (float*)partials, M, C, mpb);
}

extern "C" void launch_bn_bwd_grads(const void* partials, int msplit,
                                    void* bsums, void* dgamma, void* dbeta,
                                    int C, hipStream_t stream) {
    k_bn_bwd_grads<<<C / 64, 256, 0, stream>>>(
        (const float*)partials, msplit, (float*)bsums, (float*)dgamma,
        (float*)dbeta, C);
}

extern "C" void launch_bn_bwd_apply(const void* dz, const void* x,
                                    const void* work, const void* bsums,
                                    void* dx, int64_t M, int C,
                                    hipStream_t stream) {
    const int grid = ew_grid(M * C / 8, 256, 4);
    k_bn_bwd_apply<<<grid, 256, 0, stream>>>(
        (const uint16_t*)dz, (const uint16_t*)x, (const float*)work,
        (const float*)bsums, (uint16_t*)dx, M, C);
}
