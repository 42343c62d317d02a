// Copyright (c) Flashy-AMD authors.
// NHWC bf16 implicit-GEMM convolution FORWARD for gfx950.
//
// GEMM view:  Y[M][K] = A[M][rsc] * B[rsc][K]
//   M = N*Ho*Wo (output pixels), rsc = R*S*C (filter taps, c innermost),
//   A = implicit im2col of the input (built on the fly into LDS),
//   B = weights in [K][R][S][C] layout (= B^T: each MFMA B-fragment lane
//       reads 8 consecutive rsc for its column K — 16 B contiguous, L2).
//
// Schedule per 64-deep stage (two MFMA-K subchunks), double-buffered LDS,
// ONE barrier per stage, TWO staging register sets: stage j's global
// loads are issued at step j-3 and written to LDS at step j-1, so HBM
// latency hides under two full stages; B fragments load at use (the
// weight panel is L2-resident).  Tap indices (r,s,c) advance
// INCREMENTALLY (+64 with carry) — no integer divisions steady-state.
//
// Tile template: BM in {128, 64, 32} x BN in {64, 128}, 4 waves.  Smaller
// BM keeps the deep ResNet layers (M as small as 1024) above ~208
// workgroups; BN=128 halves A re-reads when K % 128 == 0 and the grid
// stays full.  Split-K (grid.z stage slices + fp32 combine) kicks in for
// underfilled grids and for long reductions (ops/__init__.py).
// Requires: C % 8 == 0, K % 64 == 0, rsc % 32 == 0 (ResNet bodies; the
// C=3 stem has its own direct kernels below).

#include "conv_common.h"

template <int BM, bool RELU, bool SPLITK, int SUBS = 2, int BN = CONV_BN>
__global__ void __launch_bounds__(CONV_THREADS)
k_conv_fwd(const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
           uint16_t* __restrict__ y, float* __restrict__ ws_out,
           float* __restrict__ bn_ws, ConvDims d, int stages_per_split) {
    constexpr int WAVES_M = BM >= 64 ? 2 : 1;
    constexpr int WAVES_N = 4 / WAVES_M;
    constexpr int MF = BM / WAVES_M / 16;      // m fragments per wave
    constexpr int NF = BN / WAVES_N / 16;      // n fragments per wave
    constexpr int BK2 = SUBS * CONV_BK;        // stage depth (SUBS x 32)
    constexpr int CHUNKS = BM * (BK2 / 8);
    constexpr int CPT = (CHUNKS + CONV_THREADS - 1) / CONV_THREADS;

    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_m = WAVES_M == 1 ? 0 : (wid >> 1);
    const int wave_n = WAVES_M == 1 ? wid : (wid & 1);
    // XCD-aware tile order: blocks dispatch round-robin over the 8 XCDs,
    // so giving block b the tile (b&7)*(grid/8)+(b>>3) makes each XCD's
    // blocks work on CONSECUTIVE m-tiles — neighbouring tiles share input
    // halo rows, which then hit that XCD's own L2.
    unsigned bx = blockIdx.x;
    if ((gridDim.x & 7) == 0) bx = (bx & 7) * (gridDim.x >> 3) + (bx >> 3);
    const int64_t m0 = (int64_t)bx * BM;
    const int col0 = blockIdx.y * BN;

    // [sub][row][APITCH] per buffer: 80 B rows (pitch 40) = aligned,
    // conflict-free b128 groups
    __shared__ uint16_t A_lds[2][2 * BM * CONV_APITCH];

    // --- staging state: row geometry + incremental (r,s,c) tap walk ------
    int st_hi[CPT], st_wi[CPT], st_r[CPT], st_s[CPT], st_c[CPT];
    int64_t st_n[CPT];
#pragma unroll
    for (int t = 0; t < CPT; ++t) {
        const int chunk = tid + t * CONV_THREADS;
        const int row = chunk / (BK2 / 8);
        const int64_t m = m0 + row;
        if (chunk < CHUNKS && m < M) {
            const int hw = d.Ho * d.Wo;
            st_n[t] = m / hw;
            const int rem = (int)(m % hw);
            st_hi[t] = (rem / d.Wo) * d.stride - d.pad;
            st_wi[t] = (rem % d.Wo) * d.stride - d.pad;
        } else {
            st_n[t] = -1;
        }
        const int kk = (SPLITK ? blockIdx.z * stages_per_split * BK2 : 0) +
                       (chunk % (BK2 / 8)) * 8;   // tap offset of stage 0
        st_r[t] = kk / (d.S * d.C);
        const int sc = kk - st_r[t] * d.S * d.C;
        st_s[t] = sc / d.C;
        st_c[t] = sc - st_s[t] * d.C;
    }

    auto load_stage = [&](short8* dst) {
#pragma unroll
        for (int t = 0; t < CPT; ++t) {
            short8 v = {};
            if (st_n[t] >= 0 && st_r[t] < d.R) {
                const int hi = st_hi[t] + st_r[t];
                const int wi = st_wi[t] + st_s[t];
                if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                    v = *reinterpret_cast<const short8*>(
                        x + (((st_n[t] * d.H + hi) * d.W + wi) * (int64_t)d.C +
                             st_c[t]));
            }
            dst[t] = v;
            // advance tap by one stage (+BK2) with carries — no divides
            int c = st_c[t] + BK2;
            while (c >= d.C) {
                c -= d.C;
                if (++st_s[t] == d.S) { st_s[t] = 0; ++st_r[t]; }
            }
            st_c[t] = c;
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

    auto load_b = [&](short8 (*dst)[NF], int stage) {
#pragma unroll
        for (int sub = 0; sub < SUBS; ++sub)
#pragma unroll
            for (int nf = 0; nf < NF; ++nf)
                dst[sub][nf] = *reinterpret_cast<const short8*>(
                    w + (int64_t)(b_col + nf * 16) * rsc + stage * BK2 +
                    sub * CONV_BK + a_koff);
    };

    floatx4 acc[MF][NF] = {};
    const int all_stages = (rsc + BK2 - 1) / BK2;
    const int s0 = SPLITK ? blockIdx.z * stages_per_split : 0;
    const int n_stages = SPLITK
        ? (all_stages - s0 < stages_per_split ? all_stages - s0 : stages_per_split)
        : all_stages;
    // TWO staging register sets: loads for stage j land in set[j&1] and are
    // written to LDS two steps later — global latency hides under 2 full
    // stages instead of 1.
    short8 stageA[CPT], stageB[CPT];

    // prologue: s0 -> buf0 (via A); s1 -> B; s2 -> A
    load_stage(stageA);
    lds_write(A_lds[0], stageA);
    if (n_stages > 1) load_stage(stageB);
    if (n_stages > 2) load_stage(stageA);
    __syncthreads();

    // even/odd bodies keep LDS-buffer and register-set parity COMPILE-TIME
    // (a runtime [i&1] index lowers to cndmask ladders — rule 20).
    // B fragments load at use: the weight panel is L2-resident.
    auto step = [&](int i, const uint16_t* buf, uint16_t* nbuf,
                    short8 (&rset)[CPT]) {
        short8 b[SUBS][NF];
        load_b(b, s0 + i);
        if (i + 1 < n_stages) {
            lds_write(nbuf, rset);           // stage i+1
            if (i + 3 < n_stages) load_stage(rset);  // stage i+3, same set
        }
        const int kc = (s0 + i) * BK2;
#pragma unroll
        for (int sub = 0; sub < SUBS; ++sub) {
            if (kc + sub * CONV_BK >= rsc) break;
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
                    float v = acc[mf][nf][rr];
                    if (SPLITK) {
                        ws_out[((int64_t)blockIdx.z * M + row) * d.K +
                               out_col0 + nf * 16] = v;
                    } else {
                        if (RELU) v = fmaxf(v, 0.f);
                        y[row * d.K + out_col0 + nf * 16] = f32_to_bf16(v);
                    }
                }
            }

    // Fused BatchNorm statistics: per-column sum/sumsq of this block's
    // output tile -> partials[2][K][gridDim.x] at slice blockIdx.x (the
    // same layout bn_finalize combines; out-of-range rows stage zeros so
    // they contribute nothing).  Saves the bn_stats re-read of y.
    if (!SPLITK && bn_ws != nullptr) {
        float* sred = reinterpret_cast<float*>(A_lds);  // tile done: reuse
#pragma unroll
        for (int nf = 0; nf < NF; ++nf) {
            float s = 0.f, s2 = 0.f;
#pragma unroll
            for (int mf = 0; mf < MF; ++mf)
#pragma unroll
                for (int rr = 0; rr < 4; ++rr) {
                    float v = acc[mf][nf][rr];
                    if (RELU) v = fmaxf(v, 0.f);
                    s += v;
                    s2 = fmaf(v, v, s2);
                }
            s += __shfl_xor(s, 16, 64);
            s += __shfl_xor(s, 32, 64);
            s2 += __shfl_xor(s2, 16, 64);
            s2 += __shfl_xor(s2, 32, 64);
            if (lane < 16) {
                const int colL = wave_n * (BN / WAVES_N) + nf * 16 + lane;
                sred[wave_m * BN + colL] = s;
                sred[(WAVES_M + wave_m) * BN + colL] = s2;
            }
        }
        __syncthreads();
        if (tid < BN) {
            float s = 0.f, s2 = 0.f;
#pragma unroll
            for (int wm = 0; wm < WAVES_M; ++wm) {
                s += sred[wm * BN + tid];
                s2 += sred[(WAVES_M + wm) * BN + tid];
            }
            const int c = col0 + tid;
            bn_ws[((int64_t)c) * gridDim.x + blockIdx.x] = s;
            bn_ws[((int64_t)d.K + c) * gridDim.x + blockIdx.x] = s2;
        }
    }
}

// combine split-K fp32 partials -> bf16 (+optional relu)
__global__ void __launch_bounds__(256)
k_splitk_combine(const float* __restrict__ ws, uint16_t* __restrict__ out,
                 int64_t total, int zn, int relu) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i * 4 < total; i += stride) {
        float4 acc = *reinterpret_cast<const float4*>(ws + i * 4);
        for (int z = 1; z < zn; ++z) {
            const float4 v = *reinterpret_cast<const float4*>(
                ws + (int64_t)z * total + i * 4);
            acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
        }
        if (relu) {
            acc.x = fmaxf(acc.x, 0.f); acc.y = fmaxf(acc.y, 0.f);
            acc.z = fmaxf(acc.z, 0.f); acc.w = fmaxf(acc.w, 0.f);
        }
        ushort4 o;
        o.x = f32_to_bf16(acc.x); o.y = f32_to_bf16(acc.y);
        o.z = f32_to_bf16(acc.z); o.w = f32_to_bf16(acc.w);
        *reinterpret_cast<ushort4*>(out + i * 4) = o;
    }
}

extern "C" void launch_splitk_combine(const void* ws, void* out, int64_t total,
                                      int zn, int relu, hipStream_t stream) {
    k_splitk_combine<<<ew_grid(total / 4, 256, 2), 256, 0, stream>>>(
        (const float*)ws, (uint16_t*)out, total, zn, relu);
}

// pick (BM, BN) so the grid keeps >= ~208 workgroups (256 CUs); BN=128
// halves the A-tile re-reads across column tiles when K allows it.
static void pick_tile(int64_t M, int K, int* bm_out, int* bn_out) {
    if (K % 128 == 0 && (M + 127) / 128 * (K / 128) >= 208) {
        *bm_out = 128;
        *bn_out = 128;
        return;
    }
    *bn_out = 64;
    for (int bm : {128, 64}) {
        if ((M + bm - 1) / bm * (K / 64) >= 208) { *bm_out = bm; return; }
    }
    *bm_out = 32;
}

#include <cstdlib>
static int conv_subs() {   // A/B switch: FLASHY_CONV_SUBS=1 -> 32-deep stages
    static int v = [] {
        const char* e = getenv("FLASHY_CONV_SUBS");
        return (e && e[0] == '1') ? 1 : 2;
    }();
    return v;
}

// 8-wave 256-row kernel (conv_fwd8.hip) — used for the large-M layers.
extern "C" int conv_fwd8_plan(ConvDims d, int* bn_out);
extern "C" void launch_conv_fwd8(const void* x, const void* w, void* y,
                                 ConvDims d, int relu, void* bn_ws, int bn,
                                 int mtiles, hipStream_t stream);
extern "C" int conv1x1_mloop_plan(ConvDims d, int* bn_out, int* gridx_out);
extern "C" int conv_dedup_plan(ConvDims d, int* bn_out, int* tpi_out,
                               int* rows_out, int* elems_out);
extern "C" void launch_conv_dedup(const void* x, const void* w, void* y,
                                  ConvDims d, int relu, void* bn_ws, int bn,
                                  int tpi, int rows, int lds_elems,
                                  hipStream_t stream);
extern "C" void launch_conv1x1_mloop(const void* x, const void* w, void* y,
                                     ConvDims d, int relu, void* bn_ws,
                                     int bn, int gridx, int mtiles,
                                     hipStream_t stream);

// grid.x the fwd launcher will use for these dims (= the msplit of the
// fused BN-stats partials); Python sizes the partials buffer with this.
extern "C" int conv_fwd_msplit(ConvDims d) {
    int bn8, gx;
    {
        int tpi, rows, elems;
        const int gd = conv_dedup_plan(d, &bn8, &tpi, &rows, &elems);
        if (gd) return gd;
    }
    const int mtl = conv1x1_mloop_plan(d, &bn8, &gx);
    if (mtl) return 2 * mtl;   // per-(tile, wave_m) slices
    const int mt8 = conv_fwd8_plan(d, &bn8);
    if (mt8) return mt8;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    int bm, bn;
    pick_tile(M, d.K, &bm, &bn);
    return (int)((M + bm - 1) / bm);
}

extern "C" void launch_conv_fwd(const void* x, const void* w, void* y,
                                ConvDims d, int relu, void* bn_ws,
                                hipStream_t stream) {
    int bn8, gx;
    {
        int tpi, rows, elems;
        const int gd = conv_dedup_plan(d, &bn8, &tpi, &rows, &elems);
        if (gd) {
            launch_conv_dedup(x, w, y, d, relu, bn_ws, bn8, tpi, rows,
                              elems, stream);
            return;
        }
    }
    const int mtl = conv1x1_mloop_plan(d, &bn8, &gx);
    if (mtl) {
        launch_conv1x1_mloop(x, w, y, d, relu, bn_ws, bn8, gx, mtl, stream);
        return;
    }
    const int mt8 = conv_fwd8_plan(d, &bn8);
    if (mt8) {
        launch_conv_fwd8(x, w, y, d, relu, bn_ws, bn8, mt8, stream);
        return;
    }
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    int bm, bn;
    pick_tile(M, d.K, &bm, &bn);
    dim3 grid((unsigned)((M + bm - 1) / bm), (unsigned)(d.K / bn));
    if (bn == 128) {
        auto xx = (const uint16_t*)x;
        auto ww = (const uint16_t*)w;
        auto yy = (uint16_t*)y;
        if (relu)
            k_conv_fwd<128, true, false, 2, 128><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        else
            k_conv_fwd<128, false, false, 2, 128><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        return;
    }
    auto xx = (const uint16_t*)x;
    auto ww = (const uint16_t*)w;
    auto yy = (uint16_t*)y;
    if (conv_subs() == 1) {
        if (relu) {
            if (bm == 128) k_conv_fwd<128, true, false, 1><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
            else if (bm == 64) k_conv_fwd<64, true, false, 1><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
            else k_conv_fwd<32, true, false, 1><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        } else {
            if (bm == 128) k_conv_fwd<128, false, false, 1><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
            else if (bm == 64) k_conv_fwd<64, false, false, 1><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
            else k_conv_fwd<32, false, false, 1><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        }
        return;
    }
    if (relu) {
        if (bm == 128) k_conv_fwd<128, true, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        else if (bm == 64) k_conv_fwd<64, true, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        else k_conv_fwd<32, true, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
    } else {
        if (bm == 128) k_conv_fwd<128, false, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        else if (bm == 64) k_conv_fwd<64, false, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
        else k_conv_fwd<32, false, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, nullptr, (float*)bn_ws, d, 0);
    }
}

// split-K path: BM=64 tiles, grid.z over stage ranges, fp32 workspace
// [zn][M][K], then k_splitk_combine.
extern "C" void launch_conv_fwd8_splitk(const void* x, const void* w,
                                        void* ws, ConvDims d, int bn,
                                        int mtiles, int spz, int zeff,
                                        hipStream_t stream);

extern "C" void launch_conv_fwd_splitk(const void* x, const void* w, void* ws,
                                       ConvDims d, int spz, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int rsc = d.R * d.S * d.C;
    const int all_stages = (rsc + 63) / 64;
    const int zeff = (all_stages + spz - 1) / spz;
    // 8-wave split-K for the small-M long-reduction layers (r4-class):
    // same Python-decided spz/zeff and ws layout, deeper pipeline
    if (d.C % 64 == 0 && d.K % 64 == 0 && rsc % 64 == 0 &&
        (int64_t)d.N * d.H * d.W * d.C * 2 < (int64_t)0xF0000000u) {
        const int mtiles = (int)((M + 255) / 256);
        const int bn8 = d.K % 128 == 0 ? 128 : 64;
        if ((int64_t)mtiles * (d.K / bn8) * zeff >= 120) {
            launch_conv_fwd8_splitk(x, w, ws, d, bn8, mtiles, spz, zeff,
                                    stream);
            return;
        }
    }
    dim3 grid((unsigned)((M + 63) / 64), (unsigned)(d.K / CONV_BN), (unsigned)zeff);
    k_conv_fwd<64, false, true><<<grid, CONV_THREADS, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)w, nullptr, (float*)ws, nullptr,
        d, spz);
}

// ---------------------------------------------------------------------------
// Direct kernels for the C=3 stem conv (implicit-GEMM needs C%8==0).
// K must be 64 (the ResNet stem).
// ---------------------------------------------------------------------------

// MFMA stem forward:  Y[M][64] = im2col[M][rsc] * W^T[rsc][64].
// Same key fact as the stem wgrad: taps (s, c) of one filter row r are
// CONTIGUOUS x memory, so the A image stages with short loads into an
// [m][tap] LDS tile whose MFMA fragments are plain b128 reads (taps are
// the MFMA k-dim — no transpose at all here).  Weights are staged once
// per block into LDS [K][tap-padded] (pitch 168 keeps b128 fragment
// reads conflict-free); a small LDS table holds the per-tap
// (row, s, s*C+c) decode.  Tile 64(m) x 64(K) x 32(taps), <= 5 stages.
// Replaces a scalar per-pixel kernel that ran ~400 us on the 224 stem.
__global__ void __launch_bounds__(256)
k_conv_stem_fwd(const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
                uint16_t* __restrict__ y, ConvDims d) {
    constexpr int PA = 40;    // A row pitch (80 B: 16 B-aligned, no b128
    constexpr int PW = 168;   // conflicts); W row pitch (336 B, same)
    const int rsc = d.R * d.S * d.C;
    const int sC = d.S * d.C;
    const int n_stages = (rsc + 31) / 32;
    const int rsc_pad = n_stages * 32;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_m = wid >> 1;
    const int wave_n = wid & 1;
    const int64_t m0 = (int64_t)blockIdx.x * 64;

    __shared__ __attribute__((aligned(16))) uint16_t w_l[64 * PW];
    __shared__ __attribute__((aligned(16))) uint16_t A_l[2][64 * PA];
    __shared__ int tab[160];   // tap -> (roff<<16 | s<<8 | s*C+c)

    for (int i = tid; i < 64 * rsc_pad; i += 256) {
        const int k = i / rsc_pad;
        const int j = i - k * rsc_pad;
        w_l[k * PW + j] = j < rsc ? w[(int64_t)k * rsc + j] : 0;
    }
    for (int i = tid; i < rsc; i += 256) {
        const int roff = i / sC;
        const int rem = i - roff * sC;
        tab[i] = (roff << 16) | ((rem / d.C) << 8) | rem;
    }

    // this thread stages rows m_r and m_r + 32, taps oct*4 .. oct*4+3
    const int m_r = tid >> 3;
    const int oct = tid & 7;
    int64_t rn[2];
    int rhb[2], rwb[2];
    bool rok[2];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
        const int64_t m = m0 + h * 32 + m_r;
        rok[h] = m < M;
        const int64_t mm = rok[h] ? m : 0;
        const int wo = (int)(mm % d.Wo);
        const int ho = (int)((mm / d.Wo) % d.Ho);
        rn[h] = mm / ((int64_t)d.Ho * d.Wo);
        rhb[h] = ho * d.stride - d.pad;
        rwb[h] = wo * d.stride - d.pad;
    }
    __syncthreads();   // tab ready (needed by load_a)

    struct short4v { uint16_t v[4]; };
    auto load_a = [&](int st, short4v (&av)[2]) {
#pragma unroll
        for (int h = 0; h < 2; ++h)
#pragma unroll
            for (int e = 0; e < 4; ++e) {
                const int j = st * 32 + oct * 4 + e;
                uint16_t u = 0;
                if (j < rsc && rok[h]) {
                    const int t = tab[j];
                    const int hi = rhb[h] + (t >> 16);
                    const int wi = rwb[h] + ((t >> 8) & 255);
                    if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                        u = x[((rn[h] * d.H + hi) * d.W + rwb[h]) *
                              (int64_t)d.C + (t & 255)];
                }
                av[h].v[e] = u;
            }
    };
    auto write_a = [&](uint16_t (&buf)[64 * PA], const short4v (&av)[2]) {
#pragma unroll
        for (int h = 0; h < 2; ++h)
            *reinterpret_cast<short4v*>(&buf[(h * 32 + m_r) * PA + oct * 4]) =
                av[h];
    };

    const int a_row = wave_m * 32 + (lane & 15);
    const int b_col = wave_n * 32 + (lane & 15);
    const int koff = (lane >> 4) * 8;
    floatx4 acc[2][2] = {};

    short4v av[2];
    load_a(0, av);
    write_a(A_l[0], av);
    if (n_stages > 1) load_a(1, av);
    __syncthreads();

    auto step = [&](int i, const uint16_t (&buf)[64 * PA],
                    uint16_t (&nbuf)[64 * PA]) {
        if (i + 1 < n_stages) {
            write_a(nbuf, av);
            if (i + 2 < n_stages) load_a(i + 2, av);
        }
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            const short8 a = *reinterpret_cast<const short8*>(
                &buf[(a_row + mf * 16) * PA + koff]);
#pragma unroll
            for (int nf = 0; nf < 2; ++nf) {
                const short8 b = *reinterpret_cast<const short8*>(
                    &w_l[(b_col + nf * 16) * PW + i * 32 + koff]);
                acc[mf][nf] = MFMA_BF16(a, b, acc[mf][nf]);
            }
        }
        __syncthreads();
    };
    for (int i = 0; i < n_stages;) {
        step(i, A_l[0], A_l[1]);
        if (++i >= n_stages) break;
        step(i, A_l[1], A_l[0]);
        ++i;
    }

    const int64_t row0 = m0 + wave_m * 32 + (lane >> 4) * 4;
    const int col0 = wave_n * 32 + (lane & 15);
#pragma unroll
    for (int mf = 0; mf < 2; ++mf)
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
#pragma unroll
            for (int rr = 0; rr < 4; ++rr) {
                const int64_t row = row0 + mf * 16 + rr;
                if (row < M)
                    y[row * 64 + col0 + nf * 16] =
                        f32_to_bf16(acc[mf][nf][rr]);
            }
}

extern "C" void launch_conv_stem_fwd(const void* x, const void* w, void* y,
                                     ConvDims d, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    k_conv_stem_fwd<<<(unsigned)((M + 63) / 64), 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)w, (uint16_t*)y, d);
}

// Stem wgrad: dw[k][rsc] += sum_m dout[m][k] * im2col(x)[m][rsc]
// grid (K/8 octets, tap-chunks of 32, msplit).  Thread = one k of its octet
// x one of 32 m-lanes; the chunk's (r,s,c) tap decode is staged in LDS once;
// dout loads are coalesced across the octet, input taps are same-address
// broadcasts; LDS fold over m-lanes, then 8x32 atomicAdds per block.
// Any rsc (3x3x3 = 27, 7x7x3 = 147, ...).
__global__ void __launch_bounds__(256)
k_conv_stem_wgrad(const uint16_t* __restrict__ x,
                  const uint16_t* __restrict__ dout,
                  float* __restrict__ dw, ConvDims d, int m_per_block) {
    const int kl = threadIdx.x & 7;          // k within octet
    const int k = blockIdx.x * 8 + kl;
    const int mlane = threadIdx.x >> 3;      // 0..31
    const int rsc = d.R * d.S * d.C;
    const int j0 = blockIdx.y * 32;          // tap chunk base
    const int jn = min(32, rsc - j0);
    // decode the chunk base once; walk (r,s,c) incrementally per tap
    int r0 = j0 / (d.S * d.C);
    const int sc0 = j0 - r0 * d.S * d.C;
    int s0 = sc0 / d.C;
    int c0 = sc0 - s0 * d.C;

    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t ms = (int64_t)blockIdx.z * m_per_block;
    const int64_t me = min(ms + (int64_t)m_per_block, M);
    float part[32];
#pragma unroll
    for (int j = 0; j < 32; ++j) part[j] = 0.f;
    for (int64_t m = ms + mlane; m < me; m += 32) {
        const float go = bf16_to_f32(dout[m * d.K + k]);
        const int wo = (int)(m % d.Wo);
        const int ho = (int)((m / d.Wo) % d.Ho);
        const int64_t n = m / ((int64_t)d.Ho * d.Wo);
        const int hb = ho * d.stride - d.pad;
        const int wb = wo * d.stride - d.pad;
        int r = r0, sj = s0, c = c0;
        // compile-time trip count so part[j] stays register-indexed
        // (guide rule 20); `j < jn` is wave-uniform predication.
#pragma unroll
        for (int j = 0; j < 32; ++j) {
            const int hi = hb + r;
            const int wi = wb + sj;
            if (j < jn && hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                part[j] = fmaf(
                    bf16_to_f32(x[((n * d.H + hi) * d.W + wi) * (int64_t)d.C + c]),
                    go, part[j]);
            if (++c == d.C) {
                c = 0;
                if (++sj == d.S) { sj = 0; ++r; }
            }
        }
    }
    __shared__ float red[8][32][32];
#pragma unroll
    for (int j = 0; j < 32; ++j) red[kl][mlane][j] = part[j];
    __syncthreads();
    const int kk = threadIdx.x >> 5;         // 0..7
    const int jj = threadIdx.x & 31;
    if (jj < jn) {
        float acc = 0.f;
#pragma unroll 8
        for (int i = 0; i < 32; ++i) acc += red[kk][i][jj];
        if (acc != 0.f)
            atomicAdd(&dw[(int64_t)(blockIdx.x * 8 + kk) * rsc + j0 + jj], acc);
    }
}

// MFMA stem wgrad:  dW[64][taps] = dout^T[64][M] * im2col[M][taps].
// Key fact: for a FIXED filter row r, the im2col taps (s, c) of one output
// pixel are CONTIGUOUS x memory (NHWC, c innermost), so the B operand
// stages with plain loads — no gather.  Both operands land in LDS
// [m][channel] row-major and the MFMA fragments are read with
// ds_read_b64_tr_b16 (same recipe as k_conv_wgrad).  The tap tile TJ is
// padded to 32/64 columns; a chunk covers rows_per_chunk = TJ / (S*C)
// filter rows (grid.y chunks when R*S*C > TJ, e.g. the 7x7x3 stem).
// Requires K == 64, S*C <= TJ.  Replaces the scalar per-thread reduction
// (k_conv_stem_wgrad below, kept for the S*C > 64 fallback) which was
// ~6x slower (scalar-load latency bound).
template <int OFF>
__device__ inline short8 stem_tr2(const uint16_t* p) {
    union { struct { unsigned long long lo, hi; } q; short8 v; } r;
    const unsigned a = (unsigned)(unsigned long long)(const void*)p;
    asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
                 "ds_read_b64_tr_b16 %1, %2 offset:%3\n\t"
                 "s_waitcnt lgkmcnt(0)"
                 : "=v"(r.q.lo), "=v"(r.q.hi) : "v"(a), "n"(OFF));
    return r.v;
}

template <int TJ>
__global__ void __launch_bounds__(256, 4)
k_stem_wgrad_mm(const uint16_t* __restrict__ x,
                const uint16_t* __restrict__ dout, float* __restrict__ dw,
                ConvDims d, int m_per_block, int rows_per_chunk) {
    constexpr int PJ = TJ == 32 ? 36 : 72;   // pitches keep tr reads
    constexpr int PD = 72;                   // conflict-free (see wgrad)
    constexpr int EPT = TJ / 8;              // x elems per thread per 32-m
    constexpr int JF = TJ / 16;
    const int rsc = d.R * d.S * d.C;
    const int sC = d.S * d.C;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;                // wave = 16-K block
    const int r0 = blockIdx.y * rows_per_chunk;
    const int jn = (min(rows_per_chunk, d.R - r0)) * sC;
    const int jbase = r0 * sC;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t ms = (int64_t)blockIdx.z * m_per_block;
    const int64_t me = min(ms + (int64_t)m_per_block, M);

    __shared__ __attribute__((aligned(16))) uint16_t dT[2][2 * 32 * PD];
    __shared__ __attribute__((aligned(16))) uint16_t xJ[2][2 * 32 * PJ];

    const int m_r = tid >> 3;                // m row within subchunk
    const int oct = tid & 7;
    // per-elem tap decode, packed (roff<<16 | s<<8 | s*C+c) to spare VGPRs
    int e_pack[EPT];
#pragma unroll
    for (int e = 0; e < EPT; ++e) {
        const int j = oct * EPT + e;
        const int roff = j / sC;
        const int rem = j - roff * sC;
        const int ss = rem / d.C;
        e_pack[e] = (roff << 16) | (ss << 8) | rem;
    }

    auto load_stage = [&](int64_t mc, short8 (&dv)[2],
                          uint16_t (&xsv)[2][EPT]) {
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            const int64_t m = mc + sc * 32 + m_r;
            const bool ok = m < me;
            short8 v = {};
            if (ok)
                v = *reinterpret_cast<const short8*>(dout + m * d.K + oct * 8);
            dv[sc] = v;
            int64_t n = 0; int hb = 0, wb = 0;
            if (ok) {
                const int wo = (int)(m % d.Wo);
                const int ho = (int)((m / d.Wo) % d.Ho);
                n = m / ((int64_t)d.Ho * d.Wo);
                hb = ho * d.stride - d.pad;
                wb = wo * d.stride - d.pad;
            }
#pragma unroll
            for (int e = 0; e < EPT; ++e) {
                uint16_t u = 0;
                const int hi = hb + r0 + (e_pack[e] >> 16);
                const int wi = wb + ((e_pack[e] >> 8) & 255);
                // scoff = s*C + c, so the address reduces to row + wb*C + scoff
                if (ok && oct * EPT + e < jn &&
                    hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                    u = x[((n * d.H + hi) * d.W + wb) * (int64_t)d.C +
                          (e_pack[e] & 255)];
                xsv[sc][e] = u;
            }
        }
    };
    auto stage_write = [&](uint16_t (&dTb)[2 * 32 * PD],
                           uint16_t (&xJb)[2 * 32 * PJ],
                           const short8 (&dv)[2],
                           const uint16_t (&xsv)[2][EPT]) {
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            *reinterpret_cast<short8*>(
                &dTb[(sc * 32 + m_r) * PD + oct * 8]) = dv[sc];
#pragma unroll
            for (int e = 0; e < EPT; ++e)
                xJb[(sc * 32 + m_r) * PJ + oct * EPT + e] = xsv[sc][e];
        }
    };

    // tr-read per-lane gather offsets (see k_conv_wgrad for the derivation)
    const int trd = ((lane & 15) >> 2) * PD + 4 * (lane & 3) +
                    (lane >> 4) * 8 * PD;
    const int trx = ((lane & 15) >> 2) * PJ + 4 * (lane & 3) +
                    (lane >> 4) * 8 * PJ;
    floatx4 acc[JF] = {};

    const int64_t n_stages = (me - ms + 63) / 64;
    short8 dv[2];
    uint16_t xsv[2][EPT];
    load_stage(ms, dv, xsv);
    stage_write(dT[0], xJ[0], dv, xsv);
    if (n_stages > 1) load_stage(ms + 64, dv, xsv);
    __syncthreads();

    auto step = [&](int64_t i, const uint16_t (&dTb)[2 * 32 * PD],
                    const uint16_t (&xJb)[2 * 32 * PJ],
                    uint16_t (&ndT)[2 * 32 * PD],
                    uint16_t (&nxJ)[2 * 32 * PJ]) {
        if (i + 1 < n_stages) {
            stage_write(ndT, nxJ, dv, xsv);
            if (i + 2 < n_stages) load_stage(ms + (i + 2) * 64, dv, xsv);
        }
#pragma unroll
        for (int sc = 0; sc < 2; ++sc) {
            const short8 a = stem_tr2<4 * PD * 2>(
                &dTb[sc * 32 * PD + trd + wid * 16]);
#pragma unroll
            for (int jf = 0; jf < JF; ++jf) {
                const short8 b = stem_tr2<4 * PJ * 2>(
                    &xJb[sc * 32 * PJ + trx + jf * 16]);
                acc[jf] = MFMA_BF16(a, b, acc[jf]);
            }
        }
        __syncthreads();
    };
    for (int64_t i = 0; i < n_stages;) {
        step(i, dT[0], xJ[0], dT[1], xJ[1]);
        if (++i >= n_stages) break;
        step(i, dT[1], xJ[1], dT[0], xJ[0]);
        ++i;
    }

    const int k_out = wid * 16 + (lane >> 4) * 4;
    const int j_lane = lane & 15;
#pragma unroll
    for (int jf = 0; jf < JF; ++jf) {
        const int j = jf * 16 + j_lane;
        if (j < jn)
#pragma unroll
            for (int rr = 0; rr < 4; ++rr)
                atomicAdd(&dw[(int64_t)(k_out + rr) * rsc + jbase + j],
                          acc[jf][rr]);
    }
}

extern "C" void launch_conv_stem_wgrad(const void* x, const void* dout, void* dw,
                                       ConvDims d, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int rsc = d.R * d.S * d.C;
    const int sC = d.S * d.C;
    if (d.K == 64 && sC <= 64) {             // MFMA path
        const int TJ = rsc <= 32 ? 32 : 64;
        const int rpc = rsc <= 32 ? d.R : (TJ / sC < d.R ? TJ / sC : d.R);
        const int chunks = (d.R + rpc - 1) / rpc;
        int64_t msplit = (M + 127) / 128;
        const int64_t cap = 384 / chunks > 0 ? 384 / chunks : 1;
        if (msplit > cap) msplit = cap;
        if (msplit < 1) msplit = 1;
        int m_per_block = (int)((M + msplit - 1) / msplit);
        m_per_block = (m_per_block + 63) / 64 * 64;
        const int zn = (int)((M + m_per_block - 1) / m_per_block);
        dim3 grid(1, (unsigned)chunks, (unsigned)zn);
        if (TJ == 32)
            k_stem_wgrad_mm<32><<<grid, 256, 0, stream>>>(
                (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d,
                m_per_block, rpc);
        else
            k_stem_wgrad_mm<64><<<grid, 256, 0, stream>>>(
                (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d,
                m_per_block, rpc);
        return;
    }
    const int jchunks = (rsc + 31) / 32;
    int64_t msplit = (M + 511) / 512;
    const int64_t cap = 256 / jchunks > 0 ? 256 / jchunks : 1;
    if (msplit > cap) msplit = cap;
    if (msplit < 1) msplit = 1;
    const int m_per_block = (int)((M + msplit - 1) / msplit);
    dim3 grid((unsigned)(d.K / 8), (unsigned)jchunks, (unsigned)msplit);
    k_conv_stem_wgrad<<<grid, 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, m_per_block);
}

// Stem dgrad (small-C edge layers: the RGB ends of the DCGAN generator /
// discriminator).  Computes the equivalent conv's dX for C <= 8, K == 64:
//   dx[n,hi,wi,c] = sum over valid (r,s):  dout[n,ho,wo,k] * w[k,r,s,c]
// with ho = (hi+pad-r)/stride when exact.  Weights staged to LDS as fp32.
// CT = exact channel count when known (3 = RGB, 4), else 8 with runtime
// masking.  The k loop runs as 8 short8 (16 B) vector loads per tap — the
// original 64 scalar 2 B loads made the DCGAN edge-conv backward
// issue-bound at ~198 us (13.8% of the GAN step, profiles/r02j).
template <int CT>
__global__ void __launch_bounds__(256)
k_conv_stem_dgrad(const uint16_t* __restrict__ dout,
                  const uint16_t* __restrict__ w,
                  uint16_t* __restrict__ dx, ConvDims d) {
    __shared__ float w_lds[64 * 160];
    const int rsc = d.R * d.S * d.C;
    for (int i = threadIdx.x; i < 64 * rsc; i += blockDim.x)
        w_lds[i] = bf16_to_f32(w[i]);   // [k][rsc] as stored
    __syncthreads();

    const int64_t total = (int64_t)d.N * d.H * d.W;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t m = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         m < total; m += stride) {
        const int wi = (int)(m % d.W);
        const int hi = (int)((m / d.W) % d.H);
        const int64_t n = m / ((int64_t)d.H * d.W);
        float acc[CT] = {};
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
                const uint16_t* gp =
                    dout + ((n * d.Ho + ho) * d.Wo + wo) * (int64_t)d.K;
                const int base = (r * d.S + s) * d.C;
#pragma unroll
                for (int kv = 0; kv < 8; ++kv) {
                    const short8 g8 =
                        *reinterpret_cast<const short8*>(gp + kv * 8);
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const float g =
                            bf16_to_f32(((const uint16_t*)&g8)[j]);
                        const int k = kv * 8 + j;
                        // compile-time trip count keeps acc[] in
                        // registers (guide rule 20)
#pragma unroll
                        for (int c = 0; c < CT; ++c)
                            if (CT <= 4 || c < d.C)
                                acc[c] = fmaf(g, w_lds[k * rsc + base + c],
                                              acc[c]);
                    }
                }
            }
        }
#pragma unroll
        for (int c = 0; c < CT; ++c)
            if (CT <= 4 || c < d.C)
                dx[m * d.C + c] = f32_to_bf16(acc[c]);
    }
}

extern "C" void launch_conv_stem_dgrad(const void* dout, const void* w,
                                       void* dx, ConvDims d,
                                       hipStream_t stream) {
    const int64_t total = (int64_t)d.N * d.H * d.W;
    const int grid = ew_grid(total, 256, 1);
    if (d.C == 3)
        k_conv_stem_dgrad<3><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dout, (const uint16_t*)w, (uint16_t*)dx, d);
    else if (d.C == 4)
        k_conv_stem_dgrad<4><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dout, (const uint16_t*)w, (uint16_t*)dx, d);
    else
        k_conv_stem_dgrad<8><<<grid, 256, 0, stream>>>(
            (const uint16_t*)dout, (const uint16_t*)w, (uint16_t*)dx, d);
}
