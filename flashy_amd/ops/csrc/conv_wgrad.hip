// Copyright (c) Flashy-AMD authors.
// NHWC bf16 implicit-GEMM convolution BACKWARD-WEIGHTS (wgrad) for gfx950.
//
// GEMM view:  dW[K][rsc] += A'[K][M] * B'[M][rsc]
//   A' = dout^T  (dout is [M][K] NHWC),  B' = implicit im2col of x,
//   reduction over M = N*Ho*Wo, split across grid.z blocks, fp32
//   atomicAdd into the flat fp32 gradient view (accumulate semantics match
//   autograd, and zero_grad() memsets the flat buffer).
//
// Both operands arrive [m][channel]-contiguous.  They are staged into LDS
// AS-LOADED ([m][channel] rows, one b128 store per thread per tensor per
// subchunk) and the MFMA fragments — which need [channel][m] — are read with
// gfx950's ds_read_b64_tr_b16 hardware transpose (guide T10): each 16-lane
// group gathers a [4 m][16 channel] block, two reads (offset:+4 rows) build
// the 8-m-deep operand.  This replaces the previous transposed-store scheme
// whose 32 scalar ds_write_b16 per thread per stage dominated issue.
// Row pitch 72 elements: 144 B rows keep b128 stores 16 B-aligned and give
// tr reads conflict-free banks (row stride 36 dwords -> {0,36,8,44} mod 64).
//
// 64-deep m-stages (two MFMA-K subchunks) in two LDS buffers, ONE barrier
// per stage, global loads for the next stage issued under the MFMA cluster
// (same schedule as conv_fwd).  Block tile 64(K) x 64(rsc) x 64(m).
// Requires: K % 64 == 0, C % 8 == 0, rsc % 64 == 0.

#include "conv_common.h"

#define WG_P 72  // LDS row pitch in bf16 elements (144 B = 9 * 16 B)

__global__ void __launch_bounds__(CONV_THREADS, 4)  // cap at 128 VGPR
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

    // [buffer][subchunk 2][m 32][WG_P], rows as loaded from global
    __shared__ __attribute__((aligned(16))) uint16_t doutT[2][2 * 32 * WG_P];
    __shared__ __attribute__((aligned(16))) uint16_t xT[2][2 * 32 * WG_P];

    floatx4 acc[2][2] = {};

    // staging: thread -> (m row within subchunk, channel octet); each stage
    // covers 64 m = 2 subchunks of 32.
    const int m_r = tid >> 3;            // 0..31
    const int k8 = (tid & 7) * 8;        // channel octet
    // this thread's im2col tap is FIXED (j0 + k8): decode it once
    const int t_r = (j0 + k8) / (d.S * d.C);
    const int t_scc = (j0 + k8) - t_r * d.S * d.C;
    const int t_s = t_scc / d.C;
    const int t_c = t_scc - t_s * d.C;
    const int t_hoff = t_r - d.pad;      // hi = ho*stride + t_hoff
    const int t_woff = t_s - d.pad;
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
                const int wo = (int)(m % d.Wo);
                const int ho = (int)((m / d.Wo) % d.Ho);
                const int64_t n = m / ((int64_t)d.Ho * d.Wo);
                const int hi = ho * d.stride + t_hoff;
                const int wi = wo * d.stride + t_woff;
                if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                    u = *reinterpret_cast<const short8*>(
                        x + (((n * d.H + hi) * d.W + wi) * (int64_t)d.C + t_c));
            }
            xv[sc] = u;
        }
    };
    auto stage_write = [&](uint16_t (&dT)[2 * 32 * WG_P],
                           uint16_t (&xTb)[2 * 32 * WG_P],
                           const short8* dv, const short8* xv) {
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            const int at = (sc * 32 + m_r) * WG_P + k8;
            *reinterpret_cast<short8*>(&dT[at]) = dv[sc];
            *reinterpret_cast<short8*>(&xTb[at]) = xv[sc];
        }
    };

    // per-lane element offset of the tr-read gather within one subchunk
    // image: group (lane>>4) covers m rows g*8..g*8+7; input lane (lane&15)
    // supplies row (k>>2), channels ch0 + 4*(k&3) (4 contiguous bf16).
    const int tr_lane = ((lane & 15) >> 2) * WG_P + 4 * (lane & 3) +
                        (lane >> 4) * 8 * WG_P;
    union U64x2 { struct { unsigned long long lo, hi; } q; short8 v; };

    const int64_t n_stages = (me - ms + 2 * CONV_BK - 1) / (2 * CONV_BK);
    short8 dv[2], xv[2];
    load_pair(ms, dv, xv);
    stage_write(doutT[0], xT[0], dv, xv);
    if (n_stages > 1) load_pair(ms + 2 * CONV_BK, dv, xv);
    __syncthreads();

    auto step = [&](int64_t i, const uint16_t (&dT)[2 * 32 * WG_P],
                    const uint16_t (&xTb)[2 * 32 * WG_P],
                    uint16_t (&ndT)[2 * 32 * WG_P],
                    uint16_t (&nxT)[2 * 32 * WG_P]) {
        if (i + 1 < n_stages) {
            stage_write(ndT, nxT, dv, xv);
            if (i + 2 < n_stages) load_pair(ms + (i + 2) * 2 * CONV_BK, dv, xv);
        }
        // Issue BOTH subchunks' 16 transpose reads up front; a counted
        // lgkmcnt(8) releases sc0's MFMAs while sc1's reads are still in
        // flight — sc1's LDS traffic issues UNDER sc0's MFMA cluster
        // (round-1 PMC: 27% issue-stall parked on the old per-sc
        // lgkmcnt(0)).  offset:576 = +4 m rows; offset:4608 = +subchunk.
        const unsigned a0 = (unsigned)(unsigned long long)(const void*)
            &dT[tr_lane + wave_k * 32];
        const unsigned a1 = a0 + 32;                    // +16 ch * 2 B
        const unsigned b0 = (unsigned)(unsigned long long)(const void*)
            &xTb[tr_lane + wave_j * 32];
        const unsigned b1 = b0 + 32;
        U64x2 af[2][2], bf[2][2];   // [sc][frag]
        asm volatile(
            "ds_read_b64_tr_b16 %0, %16\n\t"
            "ds_read_b64_tr_b16 %1, %16 offset:576\n\t"
            "ds_read_b64_tr_b16 %2, %17\n\t"
            "ds_read_b64_tr_b16 %3, %17 offset:576\n\t"
            "ds_read_b64_tr_b16 %4, %18\n\t"
            "ds_read_b64_tr_b16 %5, %18 offset:576\n\t"
            "ds_read_b64_tr_b16 %6, %19\n\t"
            "ds_read_b64_tr_b16 %7, %19 offset:576\n\t"
            "ds_read_b64_tr_b16 %8, %16 offset:4608\n\t"
            "ds_read_b64_tr_b16 %9, %16 offset:5184\n\t"
            "ds_read_b64_tr_b16 %10, %17 offset:4608\n\t"
            "ds_read_b64_tr_b16 %11, %17 offset:5184\n\t"
            "ds_read_b64_tr_b16 %12, %18 offset:4608\n\t"
            "ds_read_b64_tr_b16 %13, %18 offset:5184\n\t"
            "ds_read_b64_tr_b16 %14, %19 offset:4608\n\t"
            "ds_read_b64_tr_b16 %15, %19 offset:5184\n\t"
            "s_waitcnt lgkmcnt(8)"
            : "=&v"(af[0][0].q.lo), "=&v"(af[0][0].q.hi),
              "=&v"(af[0][1].q.lo), "=&v"(af[0][1].q.hi),
              "=&v"(bf[0][0].q.lo), "=&v"(bf[0][0].q.hi),
              "=&v"(bf[0][1].q.lo), "=&v"(bf[0][1].q.hi),
              "=&v"(af[1][0].q.lo), "=&v"(af[1][0].q.hi),
              "=&v"(af[1][1].q.lo), "=&v"(af[1][1].q.hi),
              "=&v"(bf[1][0].q.lo), "=&v"(bf[1][0].q.hi),
              "=&v"(bf[1][1].q.lo), "=&v"(bf[1][1].q.hi)
            : "v"(a0), "v"(a1), "v"(b0), "v"(b1));
#pragma unroll
        for (int kf = 0; kf < 2; ++kf)
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
                acc[kf][jf] = MFMA_BF16(af[0][kf].v, bf[0][jf].v, acc[kf][jf]);
        asm volatile("s_waitcnt lgkmcnt(0)"
                     : "+v"(af[1][0].q.lo), "+v"(af[1][0].q.hi),
                       "+v"(af[1][1].q.lo), "+v"(af[1][1].q.hi),
                       "+v"(bf[1][0].q.lo), "+v"(bf[1][0].q.hi),
                       "+v"(bf[1][1].q.lo), "+v"(bf[1][1].q.hi));
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int kf = 0; kf < 2; ++kf)
#pragma unroll
            for (int jf = 0; jf < 2; ++jf)
                acc[kf][jf] = MFMA_BF16(af[1][kf].v, bf[1][jf].v, acc[kf][jf]);
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

// ---------------------------------------------------------------------------
// 8-wave 128(K) x 128(rsc) wgrad: double the tile edge of the kernel above,
// HALVING both operands' re-read traffic (dout re-read rsc/128 times and x
// re-read K/128 times instead of /64) — round-2 profiling put wgrad at 21%
// of the ResNet-50/224 step, bound by this on-chip traffic.  Same
// tr_b16-transpose-read scheme; pitch 144 elems (72 dwords ≡ 8 mod 64:
// conflict-free row banks); 512 threads = 8 waves as 2(K) x 4(rsc); both
// subchunks' reads issued ahead of the MFMA clusters behind counted
// lgkmcnt.  Requires K % 128 == 0, rsc % 128 == 0.
// ---------------------------------------------------------------------------

#define WG_P8 144  // bf16 elems per LDS row (128 data + 16 pad)

__global__ void __launch_bounds__(512, 2)
k_conv_wgrad8(const uint16_t* __restrict__ x, const uint16_t* __restrict__ dout,
              float* __restrict__ dw, ConvDims d, int m_per_split) {
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_k = wid >> 2;   // 0..1: 64-K half
    const int wave_j = wid & 3;    // 0..3: 32-rsc quarter
    const int k0 = blockIdx.x * 128;
    const int j0 = blockIdx.y * 128;
    const int64_t ms = (int64_t)blockIdx.z * m_per_split;
    const int64_t me = min(ms + (int64_t)m_per_split, M);

    // [buffer][sc 2][m 32][WG_P8] rows as loaded
    __shared__ __attribute__((aligned(16))) uint16_t doutT[2][2 * 32 * WG_P8];
    __shared__ __attribute__((aligned(16))) uint16_t xT[2][2 * 32 * WG_P8];

    floatx4 acc[4][2] = {};

    const int m_r = tid >> 3;            // 0..63 (both subchunks)
    const int k8 = (tid & 7) * 8;        // channel octet within 64
    // two fixed taps per thread: j0 + k8 and j0 + 64 + k8
    int t_hoff[2], t_woff[2], t_c[2];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
        const int jj = j0 + h * 64 + k8;
        const int r = jj / (d.S * d.C);
        const int scc = jj - r * d.S * d.C;
        const int s = scc / d.C;
        t_c[h] = scc - s * d.C;
        t_hoff[h] = r - d.pad;
        t_woff[h] = s - d.pad;
    }
    auto load_pair = [&](int64_t mc, short8* dv, short8* xv) {
        const int64_t m = mc + m_r;
        short8 z = {};
        dv[0] = dv[1] = xv[0] = xv[1] = z;
        if (m < me) {
            dv[0] = *reinterpret_cast<const short8*>(dout + m * d.K + k0 + k8);
            dv[1] = *reinterpret_cast<const short8*>(
                dout + m * d.K + k0 + 64 + k8);
            const int wo = (int)(m % d.Wo);
            const int ho = (int)((m / d.Wo) % d.Ho);
            const int64_t n = m / ((int64_t)d.Ho * d.Wo);
#pragma unroll
            for (int h = 0; h < 2; ++h) {
                const int hi = ho * d.stride + t_hoff[h];
                const int wi = wo * d.stride + t_woff[h];
                if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                    xv[h] = *reinterpret_cast<const short8*>(
                        x + (((n * d.H + hi) * d.W + wi) * (int64_t)d.C +
                             t_c[h]));
            }
        }
    };
    auto stage_write = [&](uint16_t (&dT)[2 * 32 * WG_P8],
                           uint16_t (&xTb)[2 * 32 * WG_P8],
                           const short8* dv, const short8* xv) {
        const int at = m_r * WG_P8 + k8;   // [sc(m_r>>5)][m_r&31] row-major
#pragma unroll
        for (int h = 0; h < 2; ++h) {
            *reinterpret_cast<short8*>(&dT[at + h * 64]) = dv[h];
            *reinterpret_cast<short8*>(&xTb[at + h * 64]) = xv[h];
        }
    };

    const int tr_lane = ((lane & 15) >> 2) * WG_P8 + 4 * (lane & 3) +
                        (lane >> 4) * 8 * WG_P8;
    union U64x2 { struct { unsigned long long lo, hi; } q; short8 v; };

    const int64_t n_stages = (me - ms + 2 * CONV_BK - 1) / (2 * CONV_BK);
    short8 dv[2], xv[2];
    load_pair(ms, dv, xv);
    stage_write(doutT[0], xT[0], dv, xv);
    if (n_stages > 1) load_pair(ms + 2 * CONV_BK, dv, xv);
    __syncthreads();

    // per-sc fragment reads: dout 4 frags (wave_k*64 + kf*16), x 2 frags
    // (wave_j*32 + jf*16); offsets: +4 m rows = 1152 B, +subchunk = 9216 B
    auto step = [&](int64_t i, const uint16_t (&dT)[2 * 32 * WG_P8],
                    const uint16_t (&xTb)[2 * 32 * WG_P8],
                    uint16_t (&ndT)[2 * 32 * WG_P8],
                    uint16_t (&nxT)[2 * 32 * WG_P8]) {
        if (i + 1 < n_stages) {
            stage_write(ndT, nxT, dv, xv);
            if (i + 2 < n_stages) load_pair(ms + (i + 2) * 2 * CONV_BK, dv, xv);
        }
        const unsigned a0 = (unsigned)(unsigned long long)(const void*)
            &dT[tr_lane + wave_k * 64];
        const unsigned b0 = (unsigned)(unsigned long long)(const void*)
            &xTb[tr_lane + wave_j * 32];
        U64x2 af[2][4], bf[2][2];   // [sc][frag]
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            // 12 reads per subchunk in one asm block (24 outs + 2 addr ins)
            asm volatile(
                "ds_read_b64_tr_b16 %0, %12\n\t"
                "ds_read_b64_tr_b16 %1, %12 offset:1152\n\t"
                "ds_read_b64_tr_b16 %2, %12 offset:32\n\t"
                "ds_read_b64_tr_b16 %3, %12 offset:1184\n\t"
                "ds_read_b64_tr_b16 %4, %12 offset:64\n\t"
                "ds_read_b64_tr_b16 %5, %12 offset:1216\n\t"
                "ds_read_b64_tr_b16 %6, %12 offset:96\n\t"
                "ds_read_b64_tr_b16 %7, %12 offset:1248\n\t"
                "ds_read_b64_tr_b16 %8, %13\n\t"
                "ds_read_b64_tr_b16 %9, %13 offset:1152\n\t"
                "ds_read_b64_tr_b16 %10, %13 offset:32\n\t"
                "ds_read_b64_tr_b16 %11, %13 offset:1184\n\t"
                : "=&v"(af[sc][0].q.lo), "=&v"(af[sc][0].q.hi),
                  "=&v"(af[sc][1].q.lo), "=&v"(af[sc][1].q.hi),
                  "=&v"(af[sc][2].q.lo), "=&v"(af[sc][2].q.hi),
                  "=&v"(af[sc][3].q.lo), "=&v"(af[sc][3].q.hi),
                  "=&v"(bf[sc][0].q.lo), "=&v"(bf[sc][0].q.hi),
                  "=&v"(bf[sc][1].q.lo), "=&v"(bf[sc][1].q.hi)
                : "v"(a0 + sc * 9216), "v"(b0 + sc * 9216));
        }
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            if (sc == 0)
                asm volatile("s_waitcnt lgkmcnt(12)"
                             : "+v"(af[0][0].q.lo), "+v"(af[0][0].q.hi),
                               "+v"(af[0][1].q.lo), "+v"(af[0][1].q.hi),
                               "+v"(af[0][2].q.lo), "+v"(af[0][2].q.hi),
                               "+v"(af[0][3].q.lo), "+v"(af[0][3].q.hi),
                               "+v"(bf[0][0].q.lo), "+v"(bf[0][0].q.hi),
                               "+v"(bf[0][1].q.lo), "+v"(bf[0][1].q.hi));
            else
                asm volatile("s_waitcnt lgkmcnt(0)"
                             : "+v"(af[1][0].q.lo), "+v"(af[1][0].q.hi),
                               "+v"(af[1][1].q.lo), "+v"(af[1][1].q.hi),
                               "+v"(af[1][2].q.lo), "+v"(af[1][2].q.hi),
                               "+v"(af[1][3].q.lo), "+v"(af[1][3].q.hi),
                               "+v"(bf[1][0].q.lo), "+v"(bf[1][0].q.hi),
                               "+v"(bf[1][1].q.lo), "+v"(bf[1][1].q.hi));
            __builtin_amdgcn_sched_barrier(0);
#pragma unroll
            for (int kf = 0; kf < 4; ++kf)
#pragma unroll
                for (int jf = 0; jf < 2; ++jf)
                    acc[kf][jf] =
                        MFMA_BF16(af[sc][kf].v, bf[sc][jf].v, acc[kf][jf]);
        }
        __syncthreads();
    };
    for (int64_t i = 0; i < n_stages;) {
        step(i, doutT[0], xT[0], doutT[1], xT[1]);
        if (++i >= n_stages) break;
        step(i, doutT[1], xT[1], doutT[0], xT[0]);
        ++i;
    }

    const int out_k0 = k0 + wave_k * 64 + (lane >> 4) * 4;
    const int out_j0 = j0 + wave_j * 32 + (lane & 15);
#pragma unroll
    for (int kf = 0; kf < 4; ++kf)
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
    // 128x128 tile only where re-read traffic dominates (R*S > 1): the 1x1
    // wgrads already run at the HBM roof on the 64x64 kernel's 4-blocks/CU
    // occupancy, and the 1-block/CU 8-wave kernel loses TLP there (measured:
    // 1x1 196->156 TF, 3x3 186->244 TF).  Within 3x3, the measured win/loss
    // split over the ResNet-50/224 + CIFAR shapes: big M always wins, small
    // M only with enough (K,rsc) tiles to fill the chip.
    const int tiles8 = (d.K / 128) * (rsc / 128);
    const bool wg8 = d.R * d.S > 1 && d.K % 128 == 0 && rsc % 128 == 0 &&
                     (M >= 32768 || (M >= 8192 && tiles8 >= 18) ||
                      tiles8 >= 100);
    if (wg8) {
        const int tiles = (d.K / 128) * (rsc / 128);
        int ns = n_splits;
        if (ns > 1) {   // re-target for 512-thread blocks (one per CU)
            ns = 512 / tiles;
            if (ns < 1) ns = 1;
            if (ns > 128) ns = 128;
            const int64_t mcap = M / 64;
            if (ns > mcap) ns = (int)(mcap ? mcap : 1);
        }
        int mps = (int)((M + ns - 1) / ns);
        mps = (mps + 2 * CONV_BK - 1) / (2 * CONV_BK) * (2 * CONV_BK);
        const int zn = (int)((M + mps - 1) / mps);
        dim3 grid((unsigned)(d.K / 128), (unsigned)(rsc / 128), (unsigned)zn);
        k_conv_wgrad8<<<grid, 512, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, mps);
        return;
    }
    int m_per_split = (int)((M + n_splits - 1) / n_splits);
    m_per_split = (m_per_split + 2 * CONV_BK - 1) / (2 * CONV_BK) * (2 * CONV_BK);
    const int zn = (int)((M + m_per_split - 1) / m_per_split);
    dim3 grid((unsigned)(d.K / 64), (unsigned)(rsc / 64), (unsigned)zn);
    k_conv_wgrad<<<grid, CONV_THREADS, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, m_per_split);
}
