// Copyright (c) Flashy-AMD authors.
// NHWC bf16 implicit-GEMM convolution BACKWARD-DATA (dgrad) for gfx950.
//
// GEMM view:  dX[M][C] = A[M][rsk] * B[rsk][C]
//   M = N*H*W (input pixels), rsk = R*S*K (k innermost),
//   A[m][(r,s,k)] = dout[n, (hi+pad-r)/stride, (wi+pad-s)/stride, k] when the
//   division is exact and in range, else 0 (zero-fill handles stride>1),
//   B[(r,s,k)][c] = w[k][r][s][c] read from the RSCK transposed copy
//   (k_weight_transpose below) so each MFMA B-fragment lane reads 8
//   consecutive k — contiguous 16 B.
// Same pipelined 64-deep double-buffered single-barrier structure as the
// forward kernel (two staging register sets, loads 3 steps ahead), with
// an incremental (r,s,k) tap walk; B fragments load at use.
// Requires: K % 64 == 0, C % 64 == 0.

#include "conv_common.h"

template <int BM, bool SPLITK, bool S1, int SUBS = 2, int BN = CONV_BN>
__global__ void __launch_bounds__(CONV_THREADS)
k_conv_dgrad(const uint16_t* __restrict__ dout, const uint16_t* __restrict__ w_rsck,
             uint16_t* __restrict__ dx, float* __restrict__ ws_out,
             ConvDims d, int stages_per_split) {
    constexpr int WAVES_M = BM >= 64 ? 2 : 1;
    constexpr int WAVES_N = 4 / WAVES_M;
    constexpr int MF = BM / WAVES_M / 16;
    constexpr int NF = BN / WAVES_N / 16;
    constexpr int BK2 = SUBS * CONV_BK;
    constexpr int CHUNKS = BM * (BK2 / 8);
    constexpr int CPT = (CHUNKS + CONV_THREADS - 1) / CONV_THREADS;

    const int rsk = d.R * d.S * d.K;
    const int64_t M = (int64_t)d.N * d.H * d.W;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_m = WAVES_M == 1 ? 0 : (wid >> 1);
    const int wave_n = WAVES_M == 1 ? wid : (wid & 1);
    unsigned bx = blockIdx.x;   // XCD-aware remap (see conv_fwd.hip)
    if ((gridDim.x & 7) == 0) bx = (bx & 7) * (gridDim.x >> 3) + (bx >> 3);
    const int64_t m0 = (int64_t)bx * BM;
    const int col0 = blockIdx.y * BN;

    __shared__ uint16_t A_lds[2][2 * BM * CONV_APITCH];

    int st_hi[CPT], st_wi[CPT], st_r[CPT], st_s[CPT], st_k[CPT];
    int64_t st_n[CPT];
#pragma unroll
    for (int t = 0; t < CPT; ++t) {
        const int chunk = tid + t * CONV_THREADS;
        const int row = chunk / (BK2 / 8);
        const int64_t m = m0 + row;
        if (chunk < CHUNKS && m < M) {
            const int hw = d.H * d.W;
            st_n[t] = m / hw;
            const int rem = (int)(m % hw);
            st_hi[t] = rem / d.W + d.pad;
            st_wi[t] = rem % d.W + d.pad;
        } else {
            st_n[t] = -1;
        }
        const int kk0 = (SPLITK ? blockIdx.z * stages_per_split * BK2 : 0) +
                        (chunk % (BK2 / 8)) * 8;
        st_r[t] = kk0 / (d.S * d.K);
        const int sk0 = kk0 - st_r[t] * d.S * d.K;
        st_s[t] = sk0 / d.K;
        st_k[t] = sk0 - st_s[t] * d.K;
    }

    auto load_stage = [&](short8* dst) {
#pragma unroll
        for (int t = 0; t < CPT; ++t) {
            short8 v = {};
            if (st_n[t] >= 0 && st_r[t] < d.R) {
                const int hnum = st_hi[t] - st_r[t];  // = ho * stride
                const int wnum = st_wi[t] - st_s[t];
                const int ho = S1 ? hnum : hnum / d.stride;
                const int wo = S1 ? wnum : wnum / d.stride;
                if (hnum >= 0 && wnum >= 0 &&
                    (S1 || (ho * d.stride == hnum && wo * d.stride == wnum)) &&
                    ho < d.Ho && wo < d.Wo)
                    v = *reinterpret_cast<const short8*>(
                        dout + (((st_n[t] * d.Ho + ho) * d.Wo + wo) * (int64_t)d.K +
                                st_k[t]));
            }
            dst[t] = v;
            int k = st_k[t] + BK2;
            while (k >= d.K) {
                k -= d.K;
                if (++st_s[t] == d.S) { st_s[t] = 0; ++st_r[t]; }
            }
            st_k[t] = k;
        }
    };
    auto lds_write = [&](uint16_t* buf, const short8* src) {
#pragma unroll
        for (int t = 0; t < CPT; ++t) {
            const int chunk = tid + t * CONV_THREADS;
            if (chunk < CHUNKS) {
                const int row = chunk / (BK2 / 8);
                const int koff = (chunk % (BK2 / 8)) * 8;
                const int sub = koff >> 5;
                *reinterpret_cast<short8*>(
                    &buf[(sub * BM + row) * CONV_APITCH + (koff & 31)]) = src[t];
            }
        }
    };

    const int a_row = wave_m * (BM / WAVES_M) + (lane & 15);
    const int a_koff = (lane >> 4) * 8;
    const int b_col = col0 + wave_n * (BN / WAVES_N) + (lane & 15);

    // B tap state: (r, s, k) of (stage base + a_koff); (r,s) constant across
    // each 32-subchunk since K % 32 == 0, so track per sub ∈ {0,1}.
    int b_r[2], b_s[2], b_k[2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
        const int kk0 = (SPLITK ? blockIdx.z * stages_per_split * BK2 : 0) +
                        sub * CONV_BK + a_koff;
        b_r[sub] = kk0 / (d.S * d.K);
        const int sk0 = kk0 - b_r[sub] * d.S * d.K;
        b_s[sub] = sk0 / d.K;
        b_k[sub] = sk0 - b_s[sub] * d.K;
    }
    auto load_b = [&](short8 (*dst)[NF]) {
#pragma unroll
        for (int sub = 0; sub < SUBS; ++sub) {
#pragma unroll
            for (int nf = 0; nf < NF; ++nf)
                dst[sub][nf] = (b_r[sub] < d.R)
                    ? *reinterpret_cast<const short8*>(
                          w_rsck + ((int64_t)(b_r[sub] * d.S + b_s[sub]) * d.C +
                                    b_col + nf * 16) * d.K + b_k[sub])
                    : short8{};
            int k = b_k[sub] + BK2;
            while (k >= d.K) {
                k -= d.K;
                if (++b_s[sub] == d.S) { b_s[sub] = 0; ++b_r[sub]; }
            }
            b_k[sub] = k;
        }
    };

    floatx4 acc[MF][NF] = {};
    const int all_stages = (rsk + BK2 - 1) / BK2;
    const int s0 = SPLITK ? blockIdx.z * stages_per_split : 0;
    const int n_stages = SPLITK
        ? (all_stages - s0 < stages_per_split ? all_stages - s0 : stages_per_split)
        : all_stages;
    short8 stageA[CPT], stageB[CPT];

    load_stage(stageA);
    lds_write(A_lds[0], stageA);
    if (n_stages > 1) load_stage(stageB);
    if (n_stages > 2) load_stage(stageA);
    __syncthreads();

    auto step = [&](int i, const uint16_t* buf, uint16_t* nbuf,
                    short8 (&rset)[CPT]) {
        short8 b[SUBS][NF];
        load_b(b);
        if (i + 1 < n_stages) {
            lds_write(nbuf, rset);
            if (i + 3 < n_stages) load_stage(rset);
        }
        const int kc = (s0 + i) * BK2;
#pragma unroll
        for (int sub = 0; sub < SUBS; ++sub) {
            if (kc + sub * CONV_BK >= rsk) break;
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const short8 a = *reinterpret_cast<const short8*>(
                    &buf[(sub * BM + a_row + mf * 16) * CONV_APITCH + a_koff]);
#pragma unroll
                for (int nf = 0; nf < NF; ++nf)
                    acc[mf][nf] = MFMA_BF16(a, b[sub][nf], acc[mf][nf]);
            }
        }
        __syncthreads();
    };
    for (int i = 0; i < n_stages;) {
        step(i, A_lds[0], A_lds[1], stageB);
        if (++i >= n_stages) break;
        step(i, A_lds[1], A_lds[0], stageA);
        ++i;
    }

    const int64_t out_row0 = m0 + wave_m * (BM / WAVES_M) + (lane >> 4) * 4;
    const int out_col0 = col0 + wave_n * (BN / WAVES_N) + (lane & 15);
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
#pragma unroll
            for (int rr = 0; rr < 4; ++rr) {
                const int64_t row = out_row0 + mf * 16 + rr;
                if (row < M) {
                    if (SPLITK)
                        ws_out[((int64_t)blockIdx.z * M + row) * d.C +
                               out_col0 + nf * 16] = acc[mf][nf][rr];
                    else
                        dx[row * d.C + out_col0 + nf * 16] =
                            f32_to_bf16(acc[mf][nf][rr]);
                }
            }
}

extern "C" int launch_conv_dgrad_s2(const void* dout, const void* w_rsck,
                                    void* dx, ConvDims d, hipStream_t stream);

extern "C" int conv_dgrad8_plan(ConvDims d, int* bn_out);
extern "C" void launch_conv_dgrad8(const void* dout, const void* w_rsck,
                                   void* dx, ConvDims d, int bn, int mtiles,
                                   hipStream_t stream);
extern "C" int conv_dgrad8_s2_plan(ConvDims d, int* bn_out);
extern "C" void launch_conv_dgrad8_s2(const void* dout, const void* w_rsck,
                                      void* dx, ConvDims d, int bn,
                                      int mtiles, hipStream_t stream);
extern "C" int conv1x1_mloop_plan(ConvDims d, int* bn_out, int* gridx_out);
extern "C" void launch_conv1x1_mloop(const void* x, const void* w, void* y,
                                     ConvDims d, int relu, void* bn_ws,
                                     int bn, int gridx, int mtiles,
                                     hipStream_t stream);

extern "C" void launch_conv_dgrad(const void* dout, const void* w_rsck,
                                  void* dx, ConvDims d, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.H * d.W;
    auto dd = (const uint16_t*)dout;
    auto ww = (const uint16_t*)w_rsck;
    auto xx = (uint16_t*)dx;
    if (d.stride == 2) {
        int bn8;   // 1x1 s2: quarter-size GEMM + sibling zero scatter
        const int mt8 = conv_dgrad8_s2_plan(d, &bn8);
        if (mt8) {
            launch_conv_dgrad8_s2(dout, w_rsck, dx, d, bn8, mt8, stream);
            return;
        }
    }
    if (d.R == 1 && d.S == 1 && d.stride == 1 && d.pad == 0 && d.K == 64) {
        // single-stage 1x1 dgrad = the fwd m-loop GEMM with remapped dims:
        // A = dy [M][64], B = w_rsck [C][64] (same [col][red] addressing
        // as the fwd weight), out = dx [M][C]
        ConvDims dd = d;
        dd.C = 64;
        dd.K = d.C;
        int bn8, gx;
        const int mtl = conv1x1_mloop_plan(dd, &bn8, &gx);
        if (mtl) {
            launch_conv1x1_mloop(dout, w_rsck, dx, dd, 0, nullptr, bn8, gx,
                                 mtl, stream);
            return;
        }
    }
    if (d.stride == 2 &&
        launch_conv_dgrad_s2(dout, w_rsck, dx, d, stream))
        return;  // parity-class form (no zero-filled MFMA work)
    {
        int bn8;
        const int mt8 = conv_dgrad8_plan(d, &bn8);
        if (mt8) {
            launch_conv_dgrad8(dout, w_rsck, dx, d, bn8, mt8, stream);
            return;
        }
    }
    if (d.C % 128 == 0 && (M + 127) / 128 * (d.C / 128) >= 208) {
        dim3 g((unsigned)((M + 127) / 128), (unsigned)(d.C / 128));
        if (d.stride == 1)
            k_conv_dgrad<128, false, true, 2, 128><<<g, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else
            k_conv_dgrad<128, false, false, 2, 128><<<g, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        return;
    }
    const int ktiles = d.C / CONV_BN;
    int bm = 32;
    for (int cand : {128, 64}) {
        if ((M + cand - 1) / cand * ktiles >= 208) { bm = cand; break; }
    }
    dim3 grid((unsigned)((M + bm - 1) / bm), (unsigned)(d.C / CONV_BN));
    extern int conv_subs_dg();
    if (conv_subs_dg() == 1 && d.stride == 1) {
        if (bm == 128) k_conv_dgrad<128, false, true, 1><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else if (bm == 64) k_conv_dgrad<64, false, true, 1><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else k_conv_dgrad<32, false, true, 1><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
    } else if (d.stride == 1) {
        if (bm == 128) k_conv_dgrad<128, false, true><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else if (bm == 64) k_conv_dgrad<64, false, true><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else k_conv_dgrad<32, false, true><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
    } else {
        if (bm == 128) k_conv_dgrad<128, false, false><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else if (bm == 64) k_conv_dgrad<64, false, false><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
        else k_conv_dgrad<32, false, false><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, nullptr, d, 0);
    }
}

extern "C" void launch_conv_dgrad8_splitk(const void* dout,
                                          const void* w_rsck, void* ws,
                                          ConvDims d, int bn, int mtiles,
                                          int spz, int zeff,
                                          hipStream_t stream);

extern "C" void launch_conv_dgrad_splitk(const void* dout, const void* w_rsck,
                                         void* ws, ConvDims d, int spz,
                                         hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.H * d.W;
    const int rsk = d.R * d.S * d.K;
    const int all_stages = (rsk + 63) / 64;
    const int zeff = (all_stages + spz - 1) / spz;
    // 8-wave split-K for the small-M long-reduction layers (r4-class)
    if (d.C % 64 == 0 && d.K % 64 == 0 && rsk % 64 == 0 &&
        (int64_t)d.N * d.Ho * d.Wo * d.K * 2 < (int64_t)0xF0000000u) {
        const int mtiles = (int)((M + 255) / 256);
        const int bn8 = d.C % 128 == 0 ? 128 : 64;
        if ((int64_t)mtiles * (d.C / bn8) * zeff >= 120) {
            launch_conv_dgrad8_splitk(dout, w_rsck, ws, d, bn8, mtiles, spz,
                                      zeff, stream);
            return;
        }
    }
    dim3 grid((unsigned)((M + 63) / 64), (unsigned)(d.C / CONV_BN), (unsigned)zeff);
    if (d.stride == 1)
        k_conv_dgrad<64, true, true><<<grid, CONV_THREADS, 0, stream>>>(
            (const uint16_t*)dout, (const uint16_t*)w_rsck, nullptr, (float*)ws,
            d, spz);
    else
        k_conv_dgrad<64, true, false><<<grid, CONV_THREADS, 0, stream>>>(
            (const uint16_t*)dout, (const uint16_t*)w_rsck, nullptr, (float*)ws,
            d, spz);
}

// ---------------------------------------------------------------------------
// Stride-2 dgrad, parity-class form: grid.z enumerates the 4 (hi%2, wi%2)
// classes; within a class every tap (r, s) with r = ph + 2r', s = pw + 2s'
// is VALID, so no MFMA work is zero-filled (the generic kernel wastes 3/4
// on stride 2).  Same 64-deep double-buffered single-barrier schedule.
// ---------------------------------------------------------------------------

template <int BM>
__global__ void __launch_bounds__(CONV_THREADS)
k_conv_dgrad_s2(const uint16_t* __restrict__ dout,
                const uint16_t* __restrict__ w_rsck,
                uint16_t* __restrict__ dx, ConvDims d) {
    constexpr int WAVES_M = BM >= 64 ? 2 : 1;
    constexpr int WAVES_N = 4 / WAVES_M;
    constexpr int MF = BM / WAVES_M / 16;
    constexpr int NF = CONV_BN / WAVES_N / 16;
    constexpr int BK2 = 64;
    constexpr int CHUNKS = BM * (BK2 / 8);
    constexpr int CPT = (CHUNKS + CONV_THREADS - 1) / CONV_THREADS;

    const int ph = blockIdx.z >> 1;      // parity of (hi + pad)
    const int pw = blockIdx.z & 1;
    // first input coordinate in this class, and class extents
    const int a0 = ((ph - d.pad) % 2 + 2) % 2;
    const int b0 = ((pw - d.pad) % 2 + 2) % 2;
    const int Hc = a0 < d.H ? (d.H - a0 + 1) / 2 : 0;
    const int Wc = b0 < d.W ? (d.W - b0 + 1) / 2 : 0;
    // valid tap counts
    const int Rp = ph < d.R ? (d.R - ph + 1) / 2 : 0;
    const int Sp = pw < d.S ? (d.S - pw + 1) / 2 : 0;
    const int rsk = Rp * Sp * d.K;
    const int64_t Mc = (int64_t)d.N * Hc * Wc;
    if (Mc == 0) return;
    if (rsk == 0) {
        // no tap reaches this parity class (e.g. 1x1 stride 2): its input
        // pixels receive ZERO gradient — write it, don't skip it.
        const int lane0 = threadIdx.x & 63;
        const int wid0 = threadIdx.x >> 6;
        const int wm = WAVES_M == 1 ? 0 : (wid0 >> 1);
        const int wn = WAVES_M == 1 ? wid0 : (wid0 & 1);
        const int64_t zr0 = (int64_t)blockIdx.x * BM +
            wm * (BM / WAVES_M) + (lane0 >> 4) * 4;
        const int zc0 = blockIdx.y * CONV_BN +
            wn * (CONV_BN / WAVES_N) + (lane0 & 15);
#pragma unroll
        for (int mf = 0; mf < MF; ++mf)
#pragma unroll
            for (int nf = 0; nf < NF; ++nf)
#pragma unroll
                for (int rr = 0; rr < 4; ++rr) {
                    const int64_t m = zr0 + mf * 16 + rr;
                    if (m < Mc) {
                        const int hw = Hc * Wc;
                        const int64_t n = m / hw;
                        const int rem = (int)(m % hw);
                        const int hi = a0 + 2 * (rem / Wc);
                        const int wi = b0 + 2 * (rem % Wc);
                        dx[((n * d.H + hi) * d.W + wi) * (int64_t)d.C +
                           zc0 + nf * 16] = 0;
                    }
                }
        return;
    }

    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_m = WAVES_M == 1 ? 0 : (wid >> 1);
    const int wave_n = WAVES_M == 1 ? wid : (wid & 1);
    const int64_t m0 = (int64_t)blockIdx.x * BM;
    const int col0 = blockIdx.y * CONV_BN;

    __shared__ uint16_t A_lds[2][2 * BM * CONV_APITCH];

    // staging geometry: class row -> (n, ho base, wo base); all taps valid
    // up to the usual Ho/Wo range checks at the borders.
    int st_hb[CPT], st_wb[CPT], st_r[CPT], st_s[CPT], st_k[CPT];
    int64_t st_n[CPT];
#pragma unroll
    for (int t = 0; t < CPT; ++t) {
        const int chunk = tid + t * CONV_THREADS;
        const int row = chunk / (BK2 / 8);
        const int64_t m = m0 + row;
        if (chunk < CHUNKS && m < Mc) {
            const int hw = Hc * Wc;
            st_n[t] = m / hw;
            const int rem = (int)(m % hw);
            const int hi = a0 + 2 * (rem / Wc);
            const int wi = b0 + 2 * (rem % Wc);
            st_hb[t] = (hi + d.pad - ph) >> 1;   // = ho when r' = 0
            st_wb[t] = (wi + d.pad - pw) >> 1;
        } else {
            st_n[t] = -1;
        }
        const int kk0 = (chunk % (BK2 / 8)) * 8;
        st_r[t] = kk0 / (Sp * d.K);
        const int sk0 = kk0 - st_r[t] * Sp * d.K;
        st_s[t] = sk0 / d.K;
        st_k[t] = sk0 - st_s[t] * d.K;
    }

    auto load_stage = [&](short8* dst) {
#pragma unroll
        for (int t = 0; t < CPT; ++t) {
            short8 v = {};
            if (st_n[t] >= 0 && st_r[t] < Rp) {
                const int ho = st_hb[t] - st_r[t];
                const int wo = st_wb[t] - st_s[t];
                if (ho >= 0 && ho < d.Ho && wo >= 0 && wo < d.Wo)
                    v = *reinterpret_cast<const short8*>(
                        dout + (((st_n[t] * d.Ho + ho) * d.Wo + wo) *
                                (int64_t)d.K + st_k[t]));
            }
            dst[t] = v;
            int k = st_k[t] + BK2;
            while (k >= d.K) {
                k -= d.K;
                if (++st_s[t] == Sp) { st_s[t] = 0; ++st_r[t]; }
            }
            st_k[t] = k;
        }
    };
    auto lds_write = [&](uint16_t* buf, const short8* src) {
#pragma unroll
        for (int t = 0; t < CPT; ++t) {
            const int chunk = tid + t * CONV_THREADS;
            if (chunk < CHUNKS) {
                const int row = chunk / (BK2 / 8);
                const int koff = (chunk % (BK2 / 8)) * 8;
                const int sub = koff >> 5;
                *reinterpret_cast<short8*>(
                    &buf[(sub * BM + row) * CONV_APITCH + (koff & 31)]) = src[t];
            }
        }
    };

    const int a_row = wave_m * (BM / WAVES_M) + (lane & 15);
    const int a_koff = (lane >> 4) * 8;
    const int b_col = col0 + wave_n * (CONV_BN / WAVES_N) + (lane & 15);

    int b_r[2], b_s[2], b_k[2];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
        const int kk0 = sub * CONV_BK + a_koff;
        b_r[sub] = kk0 / (Sp * d.K);
        const int sk0 = kk0 - b_r[sub] * Sp * d.K;
        b_s[sub] = sk0 / d.K;
        b_k[sub] = sk0 - b_s[sub] * d.K;
    }
    auto load_b = [&](short8 (*dst)[NF]) {
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
            for (int nf = 0; nf < NF; ++nf)
                dst[sub][nf] = (b_r[sub] < Rp)
                    ? *reinterpret_cast<const short8*>(
                          w_rsck + ((int64_t)((ph + 2 * b_r[sub]) * d.S +
                                              pw + 2 * b_s[sub]) * d.C +
                                    b_col + nf * 16) * d.K + b_k[sub])
                    : short8{};
            int k = b_k[sub] + BK2;
            while (k >= d.K) {
                k -= d.K;
                if (++b_s[sub] == Sp) { b_s[sub] = 0; ++b_r[sub]; }
            }
            b_k[sub] = k;
        }
    };

    floatx4 acc[MF][NF] = {};
    const int n_stages = (rsk + BK2 - 1) / BK2;
    short8 stageA[CPT], stageB[CPT];

    load_stage(stageA);
    lds_write(A_lds[0], stageA);
    if (n_stages > 1) load_stage(stageB);
    if (n_stages > 2) load_stage(stageA);
    __syncthreads();

    auto step = [&](int i, const uint16_t* buf, uint16_t* nbuf,
                    short8 (&rset)[CPT]) {
        short8 b[2][NF];
        load_b(b);
        if (i + 1 < n_stages) {
            lds_write(nbuf, rset);
            if (i + 3 < n_stages) load_stage(rset);
        }
        const int kc = i * BK2;
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
            if (kc + sub * CONV_BK >= rsk) break;
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const short8 a = *reinterpret_cast<const short8*>(
                    &buf[(sub * BM + a_row + mf * 16) * CONV_APITCH + a_koff]);
#pragma unroll
                for (int nf = 0; nf < NF; ++nf)
                    acc[mf][nf] = MFMA_BF16(a, b[sub][nf], acc[mf][nf]);
            }
        }
        __syncthreads();
    };
    for (int i = 0; i < n_stages;) {
        step(i, A_lds[0], A_lds[1], stageB);
        if (++i >= n_stages) break;
        step(i, A_lds[1], A_lds[0], stageA);
        ++i;
    }

    // epilogue: scatter rows back to stride-2 positions of dx
    const int64_t out_row0 = m0 + wave_m * (BM / WAVES_M) + (lane >> 4) * 4;
    const int out_col0 = col0 + wave_n * (CONV_BN / WAVES_N) + (lane & 15);
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
#pragma unroll
            for (int rr = 0; rr < 4; ++rr) {
                const int64_t m = out_row0 + mf * 16 + rr;
                if (m < Mc) {
                    const int hw = Hc * Wc;
                    const int64_t n = m / hw;
                    const int rem = (int)(m % hw);
                    const int hi = a0 + 2 * (rem / Wc);
                    const int wi = b0 + 2 * (rem % Wc);
                    dx[((n * d.H + hi) * d.W + wi) * (int64_t)d.C +
                       out_col0 + nf * 16] = f32_to_bf16(acc[mf][nf][rr]);
                }
            }
}

extern "C" int launch_conv_dgrad_s2(const void* dout, const void* w_rsck,
                                    void* dx, ConvDims d, hipStream_t stream) {
    if (d.stride != 2) return 0;
    if (d.R == 1 && d.S == 1) return 0;  // 1x1: the generic kernel is faster
    // class extents (max over classes) for grid sizing
    const int Hc = (d.H + 1) / 2;
    const int Wc = (d.W + 1) / 2;
    const int64_t Mc = (int64_t)d.N * Hc * Wc;
    const int ktiles = d.C / CONV_BN;
    int bm = 32;
    for (int cand : {128, 64}) {
        if ((Mc + cand - 1) / cand * ktiles * 4 >= 208) { bm = cand; break; }
    }
    if ((Mc + 31) / 32 * ktiles * 4 < 104) return 0;  // too small: caller falls back
    dim3 grid((unsigned)((Mc + bm - 1) / bm), (unsigned)ktiles, 4);
    auto dd = (const uint16_t*)dout;
    auto ww = (const uint16_t*)w_rsck;
    auto xx = (uint16_t*)dx;
    if (bm == 128) k_conv_dgrad_s2<128><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, d);
    else if (bm == 64) k_conv_dgrad_s2<64><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, d);
    else k_conv_dgrad_s2<32><<<grid, CONV_THREADS, 0, stream>>>(dd, ww, xx, d);
    return 1;
}

#include <cstdlib>
int conv_subs_dg() {
    static int v = [] {
        const char* e = getenv("FLASHY_CONV_SUBS");
        return (e && e[0] == '1') ? 1 : 2;
    }();
    return v;
}

// ---------------------------------------------------------------------------
// Weight transpose  [K][R*S*C] -> [R*S*C][K]  (bf16).
// ---------------------------------------------------------------------------

__device__ __forceinline__ void transpose_tile64(
        const uint16_t* __restrict__ w, uint16_t* __restrict__ wt,
        int K, int rsc, int k0, int j0);

__global__ void __launch_bounds__(256)
k_weight_transpose(const uint16_t* __restrict__ w, uint16_t* __restrict__ wt,
                   int K, int rsc) {
    const int tiles_j = (rsc + 63) / 64;
    const int tk = blockIdx.x / tiles_j;
    const int tj = blockIdx.x - tk * tiles_j;
    transpose_tile64(w, wt, K, rsc, tk * 64, tj * 64);
}

extern "C" void launch_weight_transpose(const void* w, void* wt, int K, int rsc,
                                        hipStream_t stream) {
    const int tiles = ((K + 63) / 64) * ((rsc + 63) / 64);
    k_weight_transpose<<<tiles, 256, 0, stream>>>(
        (const uint16_t*)w, (uint16_t*)wt, K, rsc);
}

// LDS-tiled 64x64 transpose step: both global sides are short8 (the naive
// element loop scatters 2 B across 64 cachelines per wave -> ~3% write
// efficiency; this version is bandwidth-bound).  Guarded for edge tiles.
__device__ __forceinline__ void transpose_tile64(
        const uint16_t* __restrict__ w, uint16_t* __restrict__ wt,
        int K, int rsc, int k0, int j0) {
    __shared__ __attribute__((aligned(16))) uint16_t t[64][72];
    const int tid = threadIdx.x;
    // read [k][j]: 64 rows x 8 short8; thread -> (row pair, j octet)
    const int rr = tid >> 3;            // 0..31
    const int oc = (tid & 7) * 8;       // j octet
#pragma unroll
    for (int h = 0; h < 2; ++h) {
        const int k = k0 + h * 32 + rr;
        short8 v = {};
        if (k < K && j0 + oc < rsc) {
            if (j0 + oc + 8 <= rsc)
                v = *reinterpret_cast<const short8*>(w + (int64_t)k * rsc + j0 + oc);
            else
                for (int e = 0; e < 8; ++e)
                    if (j0 + oc + e < rsc)
                        ((uint16_t*)&v)[e] = w[(int64_t)k * rsc + j0 + oc + e];
        }
        // transposed placement: element e lands at [j][k]
#pragma unroll
        for (int e = 0; e < 8; ++e)
            t[oc + e][h * 32 + rr] = ((const uint16_t*)&v)[e];
    }
    __syncthreads();
    // write [j][k]: thread -> (j row pair, k octet); short8 rows
#pragma unroll
    for (int h = 0; h < 2; ++h) {
        const int j = j0 + h * 32 + rr;
        if (j < rsc && k0 + oc < K) {
            const short8 v = *reinterpret_cast<const short8*>(&t[h * 32 + rr][oc]);
            if (k0 + oc + 8 <= K)
                *reinterpret_cast<short8*>(wt + (int64_t)j * K + k0 + oc) = v;
            else
                for (int e = 0; e < 8; ++e)
                    if (k0 + oc + e < K)
                        wt[(int64_t)j * K + k0 + oc + e] = ((const uint16_t*)&v)[e];
        }
    }
    __syncthreads();   // LDS reused by the next tile of this block
}

// Batched transpose of EVERY conv weight in one launch: meta[i] =
// {src_off, dst_off, K, rsc} (elements), src = the optimizer's flat bf16
// mirror, dst = the shared RSCK arena.  grid.y = conv index; grid.x tiles
// the largest conv and loops (grid-stride) over this conv's 64x64 tiles.
__global__ void __launch_bounds__(256)
k_weight_transpose_batched(const uint16_t* __restrict__ src_base,
                           uint16_t* __restrict__ dst_base,
                           const int* __restrict__ meta) {
    const int* mi = meta + blockIdx.y * 4;
    const uint16_t* w = src_base + mi[0];
    uint16_t* wt = dst_base + mi[1];
    const int K = mi[2];
    const int rsc = mi[3];
    const int tiles_j = (rsc + 63) / 64;
    const int tiles = ((K + 63) / 64) * tiles_j;
    for (int t = blockIdx.x; t < tiles; t += gridDim.x) {
        const int tk = t / tiles_j;
        const int tj = t - tk * tiles_j;
        transpose_tile64(w, wt, K, rsc, tk * 64, tj * 64);
    }
}

extern "C" void launch_weight_transpose_batched(const void* src, void* dst,
                                                const void* meta, int n_convs,
                                                int64_t max_elems,
                                                hipStream_t stream) {
    const int max_tiles = (int)((max_elems + 4095) / 4096);
    dim3 grid((unsigned)(max_tiles < 1 ? 1 : max_tiles), (unsigned)n_convs);
    k_weight_transpose_batched<<<grid, 256, 0, stream>>>(
        (const uint16_t*)src, (uint16_t*)dst, (const int*)meta);
}
