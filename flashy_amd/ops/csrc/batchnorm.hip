// Copyright (c) Flashy-AMD authors.
// NHWC BatchNorm (training) for gfx950, operating on [M, C] with
// M = N*H*W, bf16 activations, fp32 stats/params.
//
// forward:  stats (per-channel sum/sumsq, atomically accumulated)
//        -> finalize (tiny: mean/invstd/scale/shift + running stats)
//        -> apply (y = relu(scale*x + shift [+ res]))  [fused add+ReLU]
// backward: reduce (dz = dy * relu-mask; per-channel sum_dz, sum_dz*xhat;
//                   dz written out — it IS the residual gradient)
//        -> grads (tiny: dgamma/dbeta accumulated into flat fp32 grads)
//        -> apply (dx = scale*(dz - (sum_dz + xhat*sum_dzxhat)/M))
//
// Replaces the BatchNorm + ReLU + residual-add chains of the reference's
// ResNet workload with NHWC-native fused kernels.

#include "common.h"

// ---------------------------------------------------------------------------
// fwd 1: per-channel sum / sumsq.  Block: 256 threads = 64 channels x 4
// m-lanes; grid (C/64, msplit).  Partial cross-thread reduce via LDS, then
// one atomicAdd per channel per block.
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
k_bn_stats(const uint16_t* __restrict__ x, float* __restrict__ sums,
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
        atomicAdd(&sums[c], s);
        atomicAdd(&sums[C + c], s2);
    }
}

// ---------------------------------------------------------------------------
// fwd 2 (tiny): mean/invstd/scale/shift + running-stat update.
// work[0..C) = mean, [C..2C) = invstd, [2C..3C) = scale, [3C..4C) = shift
// ---------------------------------------------------------------------------

__global__ void k_bn_finalize(const float* __restrict__ sums,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              float* __restrict__ running_mean,
                              float* __restrict__ running_var,
                              float* __restrict__ work, int64_t M, int C,
                              float eps, float momentum, int update_running) {
    const int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    const float mean = sums[c] / (float)M;
    float var = sums[C + c] / (float)M - mean * mean;
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
// fwd 3: y = [relu](scale*x + shift [+ res]).  short8-vectorized rows.
// C % 8 == 0.
// ---------------------------------------------------------------------------

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

// ---------------------------------------------------------------------------
// bwd 1: dz = dy * (y > 0) [if relu]; per-channel sum_dz, sum_dz_xhat.
// dz is written out (it is also the gradient of the residual input).
// bsums[0..C) = sum_dz, [C..2C) = sum_dz*xhat.
// ---------------------------------------------------------------------------

template <bool RELU>
__global__ void __launch_bounds__(256)
k_bn_bwd_reduce(const uint16_t* __restrict__ dy, const uint16_t* __restrict__ y,
                const uint16_t* __restrict__ x, const float* __restrict__ work,
                uint16_t* __restrict__ dz_out, float* __restrict__ bsums,
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
        atomicAdd(&bsums[c], s);
        atomicAdd(&bsums[C + c], sx);
    }
}

// ---------------------------------------------------------------------------
// bwd 2 (tiny): dgamma += sum_dz_xhat ; dbeta += sum_dz  (flat fp32 grads)
// ---------------------------------------------------------------------------

__global__ void k_bn_bwd_grads(const float* __restrict__ bsums,
                               float* __restrict__ dgamma,
                               float* __restrict__ dbeta, int C) {
    const int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    dbeta[c] += bsums[c];
    dgamma[c] += bsums[C + c];
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
// launchers
// ---------------------------------------------------------------------------

static int bn_msplit(int64_t M, int C, int* m_per_block) {
    // target ~1024 blocks total
    int per = (int)((M * (C / 64) + 1023) / 1024);
    int blocks_per_col = (int)((M + per - 1) / (per > 0 ? per : 1));
    if (blocks_per_col < 1) blocks_per_col = 1;
    if (blocks_per_col > 1024) blocks_per_col = 1024;
    *m_per_block = (int)((M + blocks_per_col - 1) / blocks_per_col);
    return blocks_per_col;
}

extern "C" void launch_bn_stats(const void* x, void* sums, int64_t M, int C,
                                hipStream_t stream) {
    int mpb;
    const int msplit = bn_msplit(M, C, &mpb);
    dim3 grid((unsigned)(C / 64), (unsigned)msplit);
    k_bn_stats<<<grid, 256, 0, stream>>>((const uint16_t*)x, (float*)sums, M, C, mpb);
}

extern "C" void launch_bn_finalize(const void* sums, const void* gamma,
                                   const void* beta, void* running_mean,
                                   void* running_var, void* work, int64_t M,
                                   int C, float eps, float momentum,
                                   int update_running, hipStream_t stream) {
    const int block = 256;
    k_bn_finalize<<<(C + block - 1) / block, block, 0, stream>>>(
        (const float*)sums, (const float*)gamma, (const float*)beta,
        (float*)running_mean, (float*)running_var, (float*)work, M, C, eps,
        momentum, update_running);
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
                                     void* dz_out, void* bsums, int64_t M,
                                     int C, int relu, hipStream_t stream) {
    int mpb;
    const int msplit = bn_msplit(M, C, &mpb);
    dim3 grid((unsigned)(C / 64), (unsigned)msplit);
    if (relu)
        k_bn_bwd_reduce<true><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dy, (const uint16_t*)y, (const uint16_t*)x,
            (const float*)work, (uint16_t*)dz_out, (float*)bsums, M, C, mpb);
    else
        k_bn_bwd_reduce<false><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dy, (const uint16_t*)y, (const uint16_t*)x,
            (const float*)work, (uint16_t*)dz_out, (float*)bsums, M, C, mpb);
}

extern "C" void launch_bn_bwd_grads(const void* bsums, void* dgamma,
                                    void* dbeta, int C, hipStream_t stream) {
    const int block = 256;
    k_bn_bwd_grads<<<(C + block - 1) / block, block, 0, stream>>>(
        (const float*)bsums, (float*)dgamma, (float*)dbeta, C);
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
