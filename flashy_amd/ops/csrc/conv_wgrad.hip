// Copyright (c) Flashy-AMD authors.
// NHWC bf16 implicit-GEMM convolution BACKWARD-WEIGHTS (wgrad) for gfx950.
//
// GEMM view:  dW[K][rsc] += A'[K][M] * B'[M][rsc]
//   A' = dout^T  (dout is [M][K] NHWC),  B' = implicit im2col of x,
//   reduction over M = N*Ho*Wo, split across grid.z blocks, fp32
//   atomicAdd into the flat fp32 gradient view (accumulate semantics match
//   autograd, and zero_grad() memsets the flat buffer).
// Both operands arrive [m][channel]-contiguous, so 16 B loads stage them
// into LDS *transposed* ([channel][m]) for the MFMA fragment reads.
//
// 64-deep m-stages (two MFMA-K subchunks) in two LDS buffers, ONE barrier
// per stage, global loads for the next stage issued under the MFMA cluster
// (same schedule as conv_fwd).  Block tile 64(K) x 64(rsc) x 64(m).
// Requires: K % 64 == 0, C % 8 == 0, rsc % 64 == 0.

#include "conv_common.h"

#define WG_MP 40  // LDS m-pitch per 32-m subchunk (80 B, 16B-aligned reads)

__global__ void __launch_bounds__(CONV_THREADS)
k_conv_wgrad(const uint16_t* __restrict__ x, const uint16_t* __restrict__ dout,
             float* __restrict__ dw, ConvDims d, int m_per_split) {
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_k = wid >> 1;   // K halves (32 rows each)
    const int wave_j = wid & 1;    // rsc halves
    const int k0 = blockIdx.x * 64;
    const int j0 = blockIdx.y * 64;
    const int64_t ms = (int64_t)blockIdx.z * m_per_split;
    const int64_t me = min(ms + (int64_t)m_per_split, M);

    // [buffer][subchunk][64 channels][WG_MP m]
    __shared__ uint16_t doutT[2][2 * 64 * WG_MP];
    __shared__ uint16_t xT[2][2 * 64 * WG_MP];

    floatx4 acc[2][2] = {};
    const int frag_row = wave_k * 32 + (lane & 15);   // + kf*16  (K dim)
    const int frag_col = wave_j * 32 + (lane & 15);   // + jf*16  (rsc dim)
    const int moff = (lane >> 4) * 8;

    // staging: thread -> (m row within subchunk, channel octet); each stage
    // covers 64 m = 2 subchunks of 32.
    const int m_r = tid >> 3;            // 0..31
    const int k8 = (tid & 7) * 8;        // channel octet
    auto load_pair = [&](int64_t mc, short8* dv, short8* xv) {
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            const int64_t m = mc + sc * CONV_BK + m_r;
            short8 v = {};
            if (m < me)
                v = *reinterpret_cast<const short8*>(dout + m * d.K + k0 + k8);
            dv[sc] = v;
            short8 u = {};
            if (m < me) {
                const int jj = j0 + k8;
                const int r = jj / (d.S * d.C);
                const int scc = jj - r * d.S * d.C;
                const int s = scc / d.C;
                const int c = scc - s * d.C;
                const int wo = (int)(m % d.Wo);
                const int ho = (int)((m / d.Wo) % d.Ho);
                const int64_t n = m / ((int64_t)d.Ho * d.Wo);
                const int hi = ho * d.stride + r - d.pad;
                const int wi = wo * d.stride + s - d.pad;
                if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                    u = *reinterpret_cast<const short8*>(
                        x + (((n * d.H + hi) * d.W + wi) * (int64_t)d.C + c));
            }
            xv[sc] = u;
        }
    };
    // XOR the 8-m column group by a per-row pattern so the 8 lanes sharing
    // an m_r (channel octets k8 = 0..56) land on distinct banks (was 8-way).
    auto swz_col = [](int row, int col) {
        const int g = ((row >> 2) ^ (row >> 3)) & 3;
        return (col & 7) | ((((col >> 3) ^ g) & 3) << 3);
    };
    auto stage_write = [&](uint16_t (&dT)[2 * 64 * WG_MP],
                           uint16_t (&xTb)[2 * 64 * WG_MP],
                           const short8* dv, const short8* xv) {
#pragma unroll
        for (int sc = 0; sc < 2; ++sc)
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int row = sc * 64 + k8 + j;
                const int col = swz_col(k8 + j, m_r);
                dT[row * WG_MP + col] = ((const uint16_t*)&dv[sc])[j];
                xTb[row * WG_MP + col] = ((const uint16_t*)&xv[sc])[j];
            }
    };

    const int64_t n_stages = (me - ms + 2 * CONV_BK - 1) / (2 * CONV_BK);
    short8 dv[2], xv[2];
    load_pair(ms, dv, xv);
    stage_write(doutT[0], xT[0], dv, xv);
    if (n_stages > 1) load_pair(ms + 2 * CONV_BK, dv, xv);
    __syncthreads();

    auto step = [&](int64_t i, const uint16_t (&dT)[2 * 64 * WG_MP],
                    const uint16_t (&xTb)[2 * 64 * WG_MP],
                    uint16_t (&ndT)[2 * 64 * WG_MP],
                    uint16_t (&nxT)[2 * 64 * WG_MP]) {
        if (i + 1 < n_stages) {
            stage_write(ndT, nxT, dv, xv);
            if (i + 2 < n_stages) load_pair(ms + (i + 2) * 2 * CONV_BK, dv, xv);
        }
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            short8 a[2], b[2];
#pragma unroll
            for (int f = 0; f < 2; ++f) {
                // reads stay contiguous: the swizzle only permutes which
                // 8-m group sits at moff for this row
                a[f] = *reinterpret_cast<const short8*>(
                    &dT[(sc * 64 + frag_row + f * 16) * WG_MP +
                        swz_col(frag_row + f * 16, moff)]);
                b[f] = *reinterpret_cast<const short8*>(
                    &xTb[(sc * 64 + frag_col + f * 16) * WG_MP +
                         swz_col(frag_col + f * 16, moff)]);
            }
#pragma unroll
            for (int kf = 0; kf < 2; ++kf)
#pragma unroll
                for (int jf = 0; jf < 2; ++jf)
                    acc[kf][jf] = MFMA_BF16(a[kf], b[jf], acc[kf][jf]);
        }
        __syncthreads();
    };
    for (int64_t i = 0; i < n_stages;) {
        step(i, doutT[0], xT[0], doutT[1], xT[1]);
        if (++i >= n_stages) break;
        step(i, doutT[1], xT[1], doutT[0], xT[0]);
        ++i;
    }

    // ---- accumulate into fp32 dw (flat KRSC layout) ---------------------
    const int out_k0 = k0 + wave_k * 32 + (lane >> 4) * 4;
    const int out_j0 = j0 + wave_j * 32 + (lane & 15);
#pragma unroll
    for (int kf = 0; kf < 2; ++kf)
#pragma unroll
        for (int jf = 0; jf < 2; ++jf)
#pragma unroll
            for (int rr = 0; rr < 4; ++rr) {
                const int k = out_k0 + kf * 16 + rr;
                const int j = out_j0 + jf * 16;
                atomicAdd(&dw[(int64_t)k * rsc + j], acc[kf][jf][rr]);
            }
}

extern "C" void launch_conv_wgrad(const void* x, const void* dout, void* dw,
                                  ConvDims d, int n_splits, hipStream_t stream) {
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    int m_per_split = (int)((M + n_splits - 1) / n_splits);
    m_per_split = (m_per_split + 2 * CONV_BK - 1) / (2 * CONV_BK) * (2 * CONV_BK);
    const int zn = (int)((M + m_per_split - 1) / m_per_split);
    dim3 grid((unsigned)(d.K / 64), (unsigned)(rsc / 64), (unsigned)zn);
    k_conv_wgrad<<<grid, CONV_THREADS, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, m_per_split);
}
