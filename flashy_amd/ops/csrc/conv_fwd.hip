// Copyright (c) Flashy-AMD authors.
// NHWC bf16 implicit-GEMM convolution FORWARD for gfx950.
//
// GEMM view:  Y[M][K] = A[M][rsc] * B[rsc][K]
//   M = N*Ho*Wo (output pixels), rsc = R*S*C (filter taps, c innermost),
//   A = implicit im2col of the input (built on the fly into LDS),
//   B = weights in torch channels_last layout [K][R][S][C]  (= B^T: each
//       MFMA B-fragment lane reads 8 consecutive rsc for its column K —
//       16 B contiguous loads, L2-resident).
// Block tile 128x64, 4 waves (2x2), per-wave 64x32 = 4x2 MFMA 16x16x32
// fragments, fp32 accumulation (canonical CDNA GEMM anatomy, guide §5).
// Requires: C % 8 == 0, K % 64 == 0, rsc % 32 == 0 (ResNet bodies; the
// C=3 stem has its own direct kernel below).

#include "conv_common.h"

template <bool RELU>
__global__ void __launch_bounds__(CONV_THREADS)
k_conv_fwd(const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
           uint16_t* __restrict__ y, ConvDims d) {
    const int rsc = d.R * d.S * d.C;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int wave_m = wid >> 1;           // 0..1 (64 rows each)
    const int wave_n = wid & 1;            // 0..1 (32 cols each)
    const int64_t m0 = (int64_t)blockIdx.x * CONV_BM;
    const int col0 = blockIdx.y * CONV_BN;

    __shared__ uint16_t A_lds[CONV_BM * CONV_APITCH];

    // --- per-thread staging rows (fixed across the K loop) ---------------
    // 512 16B-chunks per tile: chunk -> (row = chunk/4, c8 = chunk%4 * 8)
    int st_row[2], st_hi[2], st_wi[2];
    int64_t st_n[2];
    for (int t = 0; t < 2; ++t) {
        const int chunk = tid + t * CONV_THREADS;
        const int row = chunk >> 2;
        st_row[t] = row;
        const int64_t m = m0 + row;
        if (m < M) {
            const int hw = d.Ho * d.Wo;
            st_n[t] = m / hw;
            const int rem = (int)(m % hw);
            st_hi[t] = (rem / d.Wo) * d.stride - d.pad;  // hi base (r=0)
            st_wi[t] = (rem % d.Wo) * d.stride - d.pad;  // wi base (s=0)
        } else {
            st_n[t] = -1;
        }
    }

    floatx4 acc[4][2] = {};

    const int a_row = wave_m * 64 + (lane & 15);      // + mf*16
    const int a_koff = (lane >> 4) * 8;
    const int b_col = col0 + wave_n * 32 + (lane & 15);  // + nf*16

    for (int kc = 0; kc < rsc; kc += CONV_BK) {
        // --- stage A tile (im2col rows) into LDS -------------------------
        for (int t = 0; t < 2; ++t) {
            const int chunk = tid + t * CONV_THREADS;
            const int c8 = (chunk & 3) * 8;
            const int kk = kc + c8;
            const int r = kk / (d.S * d.C);
            const int sc = kk - r * d.S * d.C;
            const int s = sc / d.C;
            const int c = sc - s * d.C;
            short8 v = {};
            if (st_n[t] >= 0) {
                const int hi = st_hi[t] + r;
                const int wi = st_wi[t] + s;
                if (hi >= 0 && hi < d.H && wi >= 0 && wi < d.W) {
                    const int64_t off =
                        (((st_n[t] * d.H + hi) * d.W + wi) * (int64_t)d.C + c);
                    v = *reinterpret_cast<const short8*>(x + off);
                }
            }
            *reinterpret_cast<short8*>(&A_lds[st_row[t] * CONV_APITCH + c8]) = v;
        }
        __syncthreads();

        // --- MFMA over the 32-deep chunk ---------------------------------
        short8 b[2];
#pragma unroll
        for (int nf = 0; nf < 2; ++nf)
            b[nf] = *reinterpret_cast<const short8*>(
                w + (int64_t)(b_col + nf * 16) * rsc + kc + a_koff);
#pragma unroll
        for (int mf = 0; mf < 4; ++mf) {
            const short8 a = *reinterpret_cast<const short8*>(
                &A_lds[(a_row + mf * 16) * CONV_APITCH + a_koff]);
#pragma unroll
            for (int nf = 0; nf < 2; ++nf)
                acc[mf][nf] = MFMA_BF16(a, b[nf], acc[mf][nf]);
        }
        __syncthreads();
    }

    // --- epilogue: bf16 NHWC store --------------------------------------
    const int64_t out_row0 = m0 + wave_m * 64 + (lane >> 4) * 4;
    const int out_col0 = col0 + wave_n * 32 + (lane & 15);
#pragma unroll
    for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
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
    }
}

extern "C" void launch_conv_fwd(const void* x, const void* w, void* y,
                                ConvDims d, int relu, hipStream_t stream) {
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    dim3 grid((unsigned)((M + CONV_BM - 1) / CONV_BM), (unsigned)(d.K / CONV_BN));
    if (relu)
        k_conv_fwd<true><<<grid, CONV_THREADS, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)w, (uint16_t*)y, d);
    else
        k_conv_fwd<false><<<grid, CONV_THREADS, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)w, (uint16_t*)y, d);
}

// ---------------------------------------------------------------------------
// Direct kernel for the C=3 stem conv (implicit-GEMM needs C%8==0).
// One output element per thread, grid-stride; inputs are L2-resident.
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

// ---------------------------------------------------------------------------
// Stem weight gradient: dw[k][r][s][c] = sum_m dout[m][k] * im2col(x)[m][rsc]
// One block per k; each thread partial-accumulates all RSC taps over a
// strided slice of m, wave-reduces, lane 0 atomically adds into fp32 dw.
// RSC <= 32 (3x3x3 = 27).
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(256)
k_conv_stem_wgrad(const uint16_t* __restrict__ x,
                  const uint16_t* __restrict__ dout,
                  float* __restrict__ dw, ConvDims d) {
    const int k = blockIdx.x;
    const int rsc = d.R * d.S * d.C;
    float part[32];
    for (int j = 0; j < 32; ++j) part[j] = 0.f;
    const int64_t M = (int64_t)d.N * d.Ho * d.Wo;
    for (int64_t m = threadIdx.x; m < M; m += blockDim.x) {
        const float go = bf16_to_f32(dout[m * d.K + k]);
        if (go == 0.f) continue;
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
    k_conv_stem_wgrad<<<d.K, 256, 0, stream>>>(
        (const uint16_t*)x, (const uint16_t*)dout, (float*)dw, d);
}
