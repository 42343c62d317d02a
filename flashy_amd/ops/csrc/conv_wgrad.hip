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
// Block tile 64(K) x 64(rsc) x 32(m); 4 waves as 2x2.
// Requires: K % 64 == 0, C % 8 == 0, rsc % 64 == 0.

#include "conv_common.h"

#define WG_MP 40  // LDS m-pitch (bf16 elems): 80 B, 16B-aligned reads

__global__ void __launch_bounds__(CONV_THREADS)
k_conv_wgrad(const uint16_t* __restrict__ x, const uint16_t* __restrict__ dout,
             float* __restrict__ dw, ConvDims d, int m_per_split) {
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_k = wid >> 1;   // K halves (32 rows each)... see frags
    const int wave_j = wid & 1;    // rsc halves
    const int k0 = blockIdx.x * 64;
    const int j0 = blockIdx.y * 64;
    const int64_t ms = (int64_t)blockIdx.z * m_per_split;
    const int64_t me = min(ms + (int64_t)m_per_split, M);

    // transposed chunks: [64 channels][32 m], double-buffered
    __shared__ uint16_t doutT[2][64 * WG_MP];
    __shared__ uint16_t xT[2][64 * WG_MP];

    // wave computes 32(K) x 32(rsc): 2x2 fragments of 16x16
    floatx4 acc[2][2] = {};

    const int frag_row = wave_k * 32 + (lane & 15);   // + kf*16  (K dim)
    const int frag_col = wave_j * 32 + (lane & 15);   // + jf*16  (rsc dim)
    const int moff = (lane >> 4) * 8;

    // register-prefetch pipeline: the next m-chunk's global loads are in
    // flight while this chunk's MFMAs run.
    const int m_r = tid >> 3;
    const int k8 = (tid & 7) * 8;
    auto load_dout = [&](int64_t mc) -> short8 {
        short8 v = {};
        const int64_t m = mc + m_r;
        if (m < M)
            v = *reinterpret_cast<const short8*>(dout + m * d.K + k0 + k8);
        return v;
    };
    auto load_x = [&](int64_t mc) -> short8 {
        short8 v = {};
        const int64_t m = mc + m_r;
        if (m < M) {
            const int jj = j0 + k8;
            const int r = jj / (d.S * d.C);
            const int sc = jj - r * d.S * d.C;
            const int s = sc / d.C;
            const int c = sc - s * d.C;
            const int wo = (int)(m % d.Wo);
            const int ho = (int)((m / d.Wo) % d.Ho);
            const int64_t n = m / ((int64_t)d.Ho * d.Wo);
            const int hi = ho * d.stride + r - d.pad;
            const int wi = wo * d.stride + s - d.pad;
            if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                v = *reinterpret_cast<const short8*>(
                    x + (((n * d.H + hi) * d.W + wi) * (int64_t)d.C + c));
        }
        return v;
    };

    // schedule per chunk i: write regs(i+1)->buf^1, issue loads(i+2),
    // MFMA over buf, ONE barrier.  Global latency hides under ~2 chunks.
    const int64_t n_chunks = (me - ms + CONV_BK - 1) / CONV_BK;
    short8 dv = load_dout(ms), xv = load_x(ms);
    {   // prologue: chunk 0 -> buf 0; chunk 1 -> regs
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            doutT[0][(k8 + j) * WG_MP + m_r] = ((const uint16_t*)&dv)[j];
            xT[0][(k8 + j) * WG_MP + m_r] = ((const uint16_t*)&xv)[j];
        }
        if (n_chunks > 1) {
            dv = load_dout(ms + CONV_BK);
            xv = load_x(ms + CONV_BK);
        }
        __syncthreads();
    }
    for (int64_t i = 0; i < n_chunks; ++i) {
        const int cur = (int)(i & 1);
        if (i + 1 < n_chunks) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                doutT[cur ^ 1][(k8 + j) * WG_MP + m_r] = ((const uint16_t*)&dv)[j];
                xT[cur ^ 1][(k8 + j) * WG_MP + m_r] = ((const uint16_t*)&xv)[j];
            }
            if (i + 2 < n_chunks) {
                dv = load_dout(ms + (i + 2) * CONV_BK);
                xv = load_x(ms + (i + 2) * CONV_BK);
            }
        }
        short8 a[2], b[2];
#pragma unroll
        for (int f = 0; f < 2; ++f) {
            a[f] = *reinterpret_cast<const short8*>(
                &doutT[cur][(frag_row + f * 16) * WG_MP + moff]);
            b[f] = *reinterpret_cast<const short8*>(
                &xT[cur][(frag_col + f * 16) * WG_MP + moff]);
        }
#pragma unroll
        for (int kf = 0; kf < 2; ++kf)
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
                acc[kf][jf] = MFMA_BF16(a[kf], b[jf], acc[kf][jf]);
        __syncthreads();
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
    m_per_split = (m_per_split + CONV_BK - 1) / CONV_BK * CONV_BK;
    const int zn = (int)((M + m_per_split - 1) / m_per_split);
    dim3 grid((unsigned)(d.K / 64), (unsigned)(rsc / 64), (unsigned)zn);
    k_conv_wgrad<<<grid, CONV_THREADS, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, m_per_split);
}
