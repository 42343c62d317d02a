// Copyright (c) Flashy-AMD authors.
// NHWC bf16 implicit-GEMM convolution FORWARD for gfx950.
//
// GEMM view:  Y[M][K] = A[M][rsc] * B[rsc][K]
//   M = N*Ho*Wo (output pixels), rsc = R*S*C (filter taps, c innermost),
//   A = implicit im2col of the input (built on the fly into LDS),
//   B = weights in [K][R][S][C] layout (= B^T: each MFMA B-fragment lane
//       reads 8 consecutive rsc for its column K — 16 B contiguous, L2).
//
// Pipelined staging (guide §6 G15, T14 shape): the next A chunk's global
// loads are issued right after the LDS write barrier, so HBM/L2 latency
// hides under the MFMA cluster of the current chunk.
//
// Tile template: BM in {128, 64, 32} x BN=64 x BK=32, 4 waves.  Smaller BM
// keeps the deep ResNet layers (M = N*Ho*Wo as small as 1024) above ~256
// workgroups so the 256-CU chip stays filled.
// Requires: C % 8 == 0, K % 64 == 0, rsc % 32 == 0 (ResNet bodies; the
// C=3 stem has its own direct kernels below).

#include "conv_common.h"

template <int BM, bool RELU>
__global__ void __launch_bounds__(CONV_THREADS)
k_conv_fwd(const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
           uint16_t* __restrict__ y, ConvDims d) {
    constexpr int WAVES_M = BM >= 64 ? 2 : 1;
    constexpr int WAVES_N = 4 / WAVES_M;
    constexpr int MF = BM / WAVES_M / 16;      // m fragments per wave
    constexpr int NF = CONV_BN / WAVES_N / 16; // n fragments per wave
    constexpr int BK2 = 2 * CONV_BK;           // 64-deep stage (2 MFMA-K)
    constexpr int CHUNKS = BM * (BK2 / 8);     // 16B staging chunks per stage
    constexpr int CPT = (CHUNKS + CONV_THREADS - 1) / CONV_THREADS;

    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_m = WAVES_M == 1 ? 0 : (wid >> 1);
    const int wave_n = WAVES_M == 1 ? wid : (wid & 1);
    const int64_t m0 = (int64_t)blockIdx.x * BM;
    const int col0 = blockIdx.y * CONV_BN;

    // two stage buffers, each [2 kk-subchunks][BM][APITCH] (sub-major keeps
    // the row pitch at 48 elements = conflict-free ds_read_b128 groups)
    __shared__ uint16_t A_lds[2][BM * 2 * CONV_APITCH];

    int st_row[CPT], st_hi[CPT], st_wi[CPT];
    int64_t st_n[CPT];
#pragma unroll
    for (int t = 0; t < CPT; ++t) {
        const int chunk = tid + t * CONV_THREADS;
        const int row = chunk >> 3;            // 8 chunks per row (64 kk)
        st_row[t] = row;
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
    }

    // chunk load for stage base offset kc (16 B per staged chunk)
    auto load_chunk = [&](int t, int kc) -> short8 {
        short8 v = {};
        const int chunk = tid + t * CONV_THREADS;
        const int kk = kc + (chunk & 7) * 8;
        if (st_n[t] >= 0 && kk < rsc) {
            const int r = kk / (d.S * d.C);
            const int sc = kk - r * d.S * d.C;
            const int s = sc / d.C;
            const int c = sc - s * d.C;
            const int hi = st_hi[t] + r;
            const int wi = st_wi[t] + s;
            if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W)
                v = *reinterpret_cast<const short8*>(
                    x + (((st_n[t] * d.H + hi) * d.W + wi) * (int64_t)d.C + c));
        }
        return v;
    };
    // LDS layout: [sub][row][APITCH], sub = kk/32 within the 64-deep stage
    auto lds_write = [&](uint16_t* buf, int t, short8 v) {
        const int chunk = tid + t * CONV_THREADS;
        const int koff = (chunk & 7) * 8;       // 0..56 within the 64 stage
        const int sub = koff >> 5;
        *reinterpret_cast<short8*>(
            &buf[(sub * BM + st_row[t]) * CONV_APITCH + (koff & 31)]) = v;
    };

    floatx4 acc[MF][NF] = {};
    const int a_row = wave_m * (BM / WAVES_M) + (lane & 15);
    const int a_koff = (lane >> 4) * 8;
    const int b_col = col0 + wave_n * (CONV_BN / WAVES_N) + (lane & 15);

    const int n_stages = (rsc + BK2 - 1) / BK2;
    short8 stage[CPT];
    // prologue: chunk 0 -> buf0; chunk 1 -> regs
#pragma unroll
    for (int t = 0; t < CPT; ++t) stage[t] = load_chunk(t, 0);
#pragma unroll
    for (int t = 0; t < CPT; ++t)
        if (tid + t * CONV_THREADS < CHUNKS) lds_write(A_lds[0], t, stage[t]);
    if (n_stages > 1) {
#pragma unroll
        for (int t = 0; t < CPT; ++t) stage[t] = load_chunk(t, BK2);
    }
    __syncthreads();

    for (int i = 0; i < n_stages; ++i) {
        const uint16_t* buf = A_lds[i & 1];
        // regs hold stage i+1: write them to the other buffer, then start
        // loading stage i+2 (latency hides under this stage's MFMAs)
        if (i + 1 < n_stages) {
            uint16_t* nbuf = A_lds[(i + 1) & 1];
#pragma unroll
            for (int t = 0; t < CPT; ++t)
                if (tid + t * CONV_THREADS < CHUNKS) lds_write(nbuf, t, stage[t]);
            if (i + 2 < n_stages) {
#pragma unroll
                for (int t = 0; t < CPT; ++t)
                    stage[t] = load_chunk(t, (i + 2) * BK2);
            }
        }
        const int kc = i * BK2;
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
            if (kc + sub * CONV_BK >= rsc) break;
            short8 b[NF];
#pragma unroll
            for (int nf = 0; nf < NF; ++nf)
                b[nf] = *reinterpret_cast<const short8*>(
                    w + (int64_t)(b_col + nf * 16) * rsc + kc + sub * CONV_BK + a_koff);
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const short8 a = *reinterpret_cast<const short8*>(
                    &buf[(sub * BM + a_row + mf * 16) * CONV_APITCH + a_koff]);
#pragma unroll
                for (int nf = 0; nf < NF; ++nf)
                    acc[mf][nf] = MFMA_BF16(a, b[nf], acc[mf][nf]);
            }
        }
        __syncthreads();
    }

    const int64_t out_row0 = m0 + wave_m * (BM / WAVES_M) + (lane >> 4) * 4;
    const int out_col0 = col0 + wave_n * (CONV_BN / WAVES_N) + (lane & 15);
#pragma unroll
    for (int mf = 0; mf < MF; ++mf)
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
#pragma unroll
            for (int rr = 0; rr < 4; ++rr) {
                const int64_t row = out_row0 + mf * 16 + rr;
                if (row < M) {
                    float v = acc[mf][nf][rr];
                    if (RELU) v = fmaxf(v, 0.f);
                    y[row * d.K + out_col0 + nf * 16] = f32_to_bf16(v);
                }
            }
}

// pick BM so the grid keeps >= ~208 workgroups where possible (256 CUs)
static int pick_bm(int64_t M, int K) {
    const int ktiles = K / CONV_BN;
    for (int bm : {128, 64}) {
        if ((M + bm - 1) / bm * ktiles >= 208) return bm;
    }
    return 32;
}

extern "C" void launch_conv_fwd(const void* x, const void* w, void* y,
                                ConvDims d, int relu, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int bm = pick_bm(M, d.K);
    dim3 grid((unsigned)((M + bm - 1) / bm), (unsigned)(d.K / CONV_BN));
    auto xx = (const uint16_t*)x;
    auto ww = (const uint16_t*)w;
    auto yy = (uint16_t*)y;
    if (relu) {
        if (bm == 128) k_conv_fwd<128, true><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, d);
        else if (bm == 64) k_conv_fwd<64, true><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, d);
        else k_conv_fwd<32, true><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, d);
    } else {
        if (bm == 128) k_conv_fwd<128, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, d);
        else if (bm == 64) k_conv_fwd<64, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, d);
        else k_conv_fwd<32, false><<<grid, CONV_THREADS, 0, stream>>>(xx, ww, yy, d);
    }
}

// ---------------------------------------------------------------------------
// Direct kernels for the C=3 stem conv (implicit-GEMM needs C%8==0).
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
k_conv_stem_fwd(const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
                uint16_t* __restrict__ y, ConvDims d) {
    // weights staged once per block into LDS as fp32 [rsc][K] (K = 64);
    // each thread computes one output pixel x 16 consecutive channels.
    __shared__ float w_lds[32 * 64];
    const int rsc = d.R * d.S * d.C;       // <= 32 (3x3x3 = 27)
    for (int i = threadIdx.x; i < rsc * 64; i += blockDim.x) {
        const int j = i >> 6;              // tap
        const int k = i & 63;              // channel
        w_lds[j * 64 + k] = bf16_to_f32(w[(int64_t)k * rsc + j]);
    }
    __syncthreads();

    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t total = M * 4;           // 4 channel-quads of 16
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int kq = (int)(idx & 3) * 16;
        const int64_t m = idx >> 2;
        const int wo = (int)(m % d.Wo);
        const int ho = (int)((m / d.Wo) % d.Ho);
        const int64_t n = m / ((int64_t)d.Ho * d.Wo);
        float acc[16];
#pragma unroll
        for (int t = 0; t < 16; ++t) acc[t] = 0.f;
        for (int r = 0; r < d.R; ++r) {
            const int hi = ho * d.stride + r - d.pad;
            if (hi < 0 || hi >= d.H) continue;
            for (int s = 0; s < d.S; ++s) {
                const int wi = wo * d.stride + s - d.pad;
                if (wi < 0 || wi >= d.W) continue;
                const uint16_t* xp = x + ((n * d.H + hi) * d.W + wi) * d.C;
                const float* wp = w_lds + (r * d.S + s) * d.C * 64 + kq;
                for (int c = 0; c < d.C; ++c) {
                    const float xv = bf16_to_f32(xp[c]);
#pragma unroll
                    for (int t = 0; t < 16; ++t)
                        acc[t] = fmaf(xv, wp[c * 64 + t], acc[t]);
                }
            }
        }
        short8 out[2];
#pragma unroll
        for (int t = 0; t < 16; ++t)
            ((uint16_t*)out)[t] = f32_to_bf16(acc[t]);
        *reinterpret_cast<short8*>(y + m * d.K + kq) = out[0];
        *reinterpret_cast<short8*>(y + m * d.K + kq + 8) = out[1];
    }
}

extern "C" void launch_conv_stem_fwd(const void* x, const void* w, void* y,
                                     ConvDims d, hipStream_t stream) {
    const int64_t total = (int64_t)d.N * d.Ho * d.Wo * 4;
    k_conv_stem_fwd<<<ew_grid(total, 256, 1), 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)w, (uint16_t*)y, d);
}

// Stem wgrad: dw[k][rsc] += sum_m dout[m][k] * im2col(x)[m][rsc]
// grid (K/8 octets, msplit).  Thread = one k of its octet x one of 32
// m-lanes: dout loads are 16B-coalesced across the octet, the 27 input taps
// are same-address broadcasts within the octet; LDS fold over m-lanes, then
// 8x27 atomicAdds per block.  RSC <= 32 (3x3x3 = 27).
__global__ void __launch_bounds__(256)
k_conv_stem_wgrad(const uint16_t* __restrict__ x,
                  const uint16_t* __restrict__ dout,
                  float* __restrict__ dw, ConvDims d, int m_per_block) {
    const int kl = threadIdx.x & 7;          // k within octet
    const int k = blockIdx.x * 8 + kl;
    const int mlane = threadIdx.x >> 3;      // 0..31
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t ms = (int64_t)blockIdx.y * m_per_block;
    const int64_t me = min(ms + (int64_t)m_per_block, M);
    float part[32];
#pragma unroll
    for (int j = 0; j < 32; ++j) part[j] = 0.f;
    for (int64_t m = ms + mlane; m < me; m += 32) {
        const float go = bf16_to_f32(dout[m * d.K + k]);
        const int wo = (int)(m % d.Wo);
        const int ho = (int)((m / d.Wo) % d.Ho);
        const int64_t n = m / ((int64_t)d.Ho * d.Wo);
        for (int r = 0; r < d.R; ++r) {
            const int hi = ho * d.stride + r - d.pad;
            if (hi < 0 || hi >= d.H) continue;
            for (int s = 0; s < d.S; ++s) {
                const int wi = wo * d.stride + s - d.pad;
                if (wi < 0 || wi >= d.W) continue;
                const uint16_t* xp = x + ((n * d.H + hi) * d.W + wi) * d.C;
                const int base = (r * d.S + s) * d.C;
                for (int c = 0; c < d.C; ++c)
                    part[base + c] = fmaf(bf16_to_f32(xp[c]), go, part[base + c]);
            }
        }
    }
    // fold the 32 m-lanes per k: LDS [8 k][32 lanes spread over 27 taps]
    __shared__ float red[8][32][32];
#pragma unroll
    for (int j = 0; j < 32; ++j) red[kl][mlane][j] = part[j];
    __syncthreads();
    // 256 threads: thread -> (k-octet slot, tap); fold 32 lanes
    const int kk = threadIdx.x >> 5;         // 0..7
    const int j0 = threadIdx.x & 31;         // tap (first 27 valid)
    if (j0 < rsc) {
        float acc = 0.f;
#pragma unroll 8
        for (int i = 0; i < 32; ++i) acc += red[kk][i][j0];
        if (acc != 0.f)
            atomicAdd(&dw[(int64_t)(blockIdx.x * 8 + kk) * rsc + j0], acc);
    }
}

extern "C" void launch_conv_stem_wgrad(const void* x, const void* dout, void* dw,
                                       ConvDims d, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    int64_t msplit = (M + 511) / 512;
    if (msplit > 128) msplit = 128;
    if (msplit < 1) msplit = 1;
    const int m_per_block = (int)((M + msplit - 1) / msplit);
    dim3 grid((unsigned)(d.K / 8), (unsigned)msplit);
    k_conv_stem_wgrad<<<grid, 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, m_per_block);
}
