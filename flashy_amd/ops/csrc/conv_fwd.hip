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
    constexpr int CHUNKS = BM * (CONV_BK / 8); // 16B staging chunks
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

    __shared__ uint16_t A_lds[BM * CONV_APITCH];

    int st_row[CPT], st_hi[CPT], st_wi[CPT];
    int64_t st_n[CPT];
#pragma unroll
    for (int t = 0; t < CPT; ++t) {
        const int chunk = tid + t * CONV_THREADS;
        const int row = chunk >> 2;
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

    // chunk load for reduction offset kc (16 B per staged chunk)
    auto load_chunk = [&](int t, int kc) -> short8 {
        short8 v = {};
        if (st_n[t] >= 0) {
            const int chunk = tid + t * CONV_THREADS;
            const int kk = kc + (chunk & 3) * 8;
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

    floatx4 acc[MF][NF] = {};
    const int a_row = wave_m * (BM / WAVES_M) + (lane & 15);
    const int a_koff = (lane >> 4) * 8;
    const int b_col = col0 + wave_n * (CONV_BN / WAVES_N) + (lane & 15);

    short8 stage[CPT];
#pragma unroll
    for (int t = 0; t < CPT; ++t) stage[t] = load_chunk(t, 0);

    for (int kc = 0; kc < rsc; kc += CONV_BK) {
        __syncthreads();  // previous chunk's LDS reads complete
#pragma unroll
        for (int t = 0; t < CPT; ++t)
            if (tid + t * CONV_THREADS < CHUNKS)
                *reinterpret_cast<short8*>(
                    &A_lds[st_row[t] * CONV_APITCH + ((tid + t * CONV_THREADS) & 3) * 8]) =
                    stage[t];
        __syncthreads();
        if (kc + CONV_BK < rsc) {
#pragma unroll
            for (int t = 0; t < CPT; ++t) stage[t] = load_chunk(t, kc + CONV_BK);
        }
        short8 b[NF];
#pragma unroll
        for (int nf = 0; nf < NF; ++nf)
            b[nf] = *reinterpret_cast<const short8*>(
                w + (int64_t)(b_col + nf * 16) * rsc + kc + a_koff);
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
            const short8 a = *reinterpret_cast<const short8*>(
                &A_lds[(a_row + mf * 16) * CONV_APITCH + a_koff]);
#pragma unroll
            for (int nf = 0; nf < NF; ++nf)
                acc[mf][nf] = MFMA_BF16(a, b[nf], acc[mf][nf]);
        }
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
    const int64_t total = (int64_t)d.N * d.Ho * d.Wo * d.K;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         idx < total; idx += stride) {
        const int k = (int)(idx % d.K);
        const int64_t m = idx / d.K;
        const int wo = (int)(m % d.Wo);
        const int ho = (int)((m / d.Wo) % d.Ho);
        const int64_t n = m / ((int64_t)d.Ho * d.Wo);
        float acc = 0.f;
        for (int r = 0; r < d.R; ++r) {
            const int hi = ho * d.stride + r - d.pad;
            if (hi < 0 || hi >= d.H) continue;
            for (int s = 0; s < d.S; ++s) {
                const int wi = wo * d.stride + s - d.pad;
                if (wi < 0 || wi >= d.W) continue;
                const uint16_t* xp = x + ((n * d.H + hi) * d.W + wi) * d.C;
                const uint16_t* wp = w + ((int64_t)k * d.R * d.S + r * d.S + s) * d.C;
                for (int c = 0; c < d.C; ++c)
                    acc = fmaf(bf16_to_f32(xp[c]), bf16_to_f32(wp[c]), acc);
            }
        }
        y[idx] = f32_to_bf16(acc);
    }
}

extern "C" void launch_conv_stem_fwd(const void* x, const void* w, void* y,
                                     ConvDims d, hipStream_t stream) {
    const int64_t total = (int64_t)d.N * d.Ho * d.Wo * d.K;
    k_conv_stem_fwd<<<ew_grid(total, 256, 1), 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)w, (uint16_t*)y, d);
}

// Stem wgrad: dw[k][rsc] += sum_m dout[m][k] * im2col(x)[m][rsc]
// grid (K, msplit): each block reduces its m-slice for one k, wave-reduces,
// atomics into fp32 dw.  RSC <= 32 (3x3x3 = 27).
__global__ void __launch_bounds__(256)
k_conv_stem_wgrad(const uint16_t* __restrict__ x,
                  const uint16_t* __restrict__ dout,
                  float* __restrict__ dw, ConvDims d, int m_per_block) {
    const int k = blockIdx.x;
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t ms = (int64_t)blockIdx.y * m_per_block;
    const int64_t me = min(ms + (int64_t)m_per_block, M);
    float part[32];
#pragma unroll
    for (int j = 0; j < 32; ++j) part[j] = 0.f;
    for (int64_t m = ms + threadIdx.x; m < me; m += blockDim.x) {
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
    for (int j = 0; j < rsc; ++j) {
        float v = wave_sum(part[j]);
        if ((threadIdx.x & 63) == 0 && v != 0.f)
            atomicAdd(&dw[(int64_t)k * rsc + j], v);
    }
}

extern "C" void launch_conv_stem_wgrad(const void* x, const void* dout, void* dw,
                                       ConvDims d, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t cap = (M + 255) / 256;
    const int msplit = (int)(cap < 32 ? (cap < 1 ? 1 : cap) : 32);
    const int m_per_block = (int)((M + msplit - 1) / msplit);
    dim3 grid((unsigned)d.K, (unsigned)msplit);
    k_conv_stem_wgrad<<<grid, 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d, m_per_block);
}
