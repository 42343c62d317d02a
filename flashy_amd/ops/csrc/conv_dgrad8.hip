// Copyright (c) Flashy-AMD authors.
// 8-wave 256-row implicit-GEMM conv BACKWARD-DATA (dgrad) for gfx950 —
// the same deep-pipeline glds schedule as conv_fwd8.hip applied to the
// dgrad GEMM (see conv_dgrad.hip for the GEMM view):
//
//   dX[M][C] = A[M][rsk] * B[rsk][C],  M = N*H*W,
//   A[m][(r,s,k)] = dout[n, (hi+pad-r)/stride, (wi+pad-s)/stride, k]
//     (exact-division + range check; zeros otherwise — the stride>1
//      zero-fill rides the same OOB-sentinel buffer bounds check),
//   B[(r,s,k)][c] = w_rsck[r][s][c][k]  (8 consecutive k = 16 B).
//
// 512 threads (2M x 4N waves), BM=256 x BN in {64,128} over C, BK=64,
// 3 LDS buffers, one raw barrier + counted vmcnt per stage, XOR-swizzled
// images, setprio around MFMA.  Requires K % 64 == 0, C % 64 == 0.

#include "conv_common.h"

#define OOB_SENTINEL 0xF0000000u

// SCAT2: 1x1 stride-2 dgrad as a quarter-size GEMM — dx is nonzero only at
// even (hi, wi), so A rows are the OUTPUT pixels read linearly from dy and
// the epilogue scatters each result to (2ho, 2wo) while writing the three
// odd-position siblings as zeros (no zero-filled MFMA work, no memset pass).
template <int BN, bool S1, bool SCAT2 = false, bool SPLITK = false>
__global__ void __launch_bounds__(512, 2)
k_conv_dgrad8(const uint16_t* __restrict__ dout,
              const uint16_t* __restrict__ w_rsck,
              uint16_t* __restrict__ dx, ConvDims d, unsigned dout_nbytes,
              int spz = 0) {
    constexpr int BM = 256;
    constexpr int BK = 64;
    constexpr int NF = BN / 64;
    constexpr int MF = 8;
    constexpr int A_ELEMS = BM * BK;
    constexpr int B_ELEMS = BN * BK;
    constexpr int BUF_ELEMS = A_ELEMS + B_ELEMS;
    constexpr int G = 4 + BN / 64;       // glds per wave per stage (A + B)
    // BN=64 buffers fit 4x in the 160 KiB LDS -> prefetch DEPTH 2 (two
    // stages' DMA in flight across barriers); BN=128 fits 3 -> depth 1
    constexpr int BUFS = BN == 64 ? 4 : 3;
    constexpr int DEPTH = BUFS - 2;

    const int rsk = d.R * d.S * d.K;
    const int64_t M = SCAT2 ? (int64_t)d.N * d.Ho * d.Wo
                            : (int64_t)d.N * d.H * d.W;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid_u = __builtin_amdgcn_readfirstlane(tid >> 6);
    const int wave_m = wid_u >> 2;
    const int wave_n = wid_u & 3;

    unsigned bx = blockIdx.x;
    if ((gridDim.x & 7) == 0)
        bx = (bx & 7) * (gridDim.x >> 3) + (bx >> 3);
    const int64_t m0 = (int64_t)bx * BM;
    const int col0 = blockIdx.y * BN;

    __shared__ uint16_t lds[BUFS * BUF_ELEMS];

    const auto arsrc = __builtin_amdgcn_make_buffer_rsrc(
        (void*)dout, 0, dout_nbytes, 0x00020000);

    // A: 4 dest chunks/thread; (r,s,k) walks +BK per stage.
    int a_r[4], a_s[4], a_k[4], a_hi[4], a_wi[4];
    int64_t a_n[4];
#pragma unroll
    for (int g = 0; g < 4; ++g) {
        const int chunk = g * 512 + tid;
        const int row = chunk >> 3;
        const int kc_s = (chunk & 7) ^ (row & 7);
        const int64_t m = m0 + row;
        if (SCAT2) {
            a_n[g] = m < M ? m : -1;   // linear dy row (1x1 over out grid)
        } else if (m < M) {
            const int hw = d.H * d.W;
            a_n[g] = m / hw;
            const int rem = (int)(m % hw);
            a_hi[g] = rem / d.W + d.pad;
            a_wi[g] = rem % d.W + d.pad;
        } else {
            a_n[g] = -1;
        }
        const int kk = (SPLITK ? blockIdx.z * spz * 64 : 0) + kc_s * 8;
        a_r[g] = kk / (d.S * d.K);
        const int sk = kk - a_r[g] * d.S * d.K;
        a_s[g] = sk / d.K;
        a_k[g] = sk - a_s[g] * d.K;
    }
    // B: per-chunk (r,s,k) walk over w_rsck[r][s][col][k].
    int b_r[2], b_s[2], b_k[2], b_col[2];
#pragma unroll
    for (int g = 0; g < NF; ++g) {
        const int chunk = g * 512 + tid;
        const int col = chunk >> 3;
        const int kc_s = (chunk & 7) ^ (col & 7);
        b_col[g] = col0 + col;
        const int kk = (SPLITK ? blockIdx.z * spz * 64 : 0) + kc_s * 8;
        b_r[g] = kk / (d.S * d.K);
        const int sk = kk - b_r[g] * d.S * d.K;
        b_s[g] = sk / d.K;
        b_k[g] = sk - b_s[g] * d.K;
    }

    auto issue_stage = [&](int buf) {
        uint16_t* base = lds + buf * BUF_ELEMS;
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            unsigned voff = OOB_SENTINEL;
            if (SCAT2) {
                if (a_n[g] >= 0)
                    voff = (unsigned)((a_n[g] * d.K + a_k[g]) * 2);
                a_k[g] += BK;   // 1x1: k never wraps within rsk
            } else if (a_n[g] >= 0 && a_r[g] < d.R) {
                const int hnum = a_hi[g] - a_r[g];   // = ho * stride
                const int wnum = a_wi[g] - a_s[g];
                const int ho = S1 ? hnum : hnum / d.stride;
                const int wo = S1 ? wnum : wnum / d.stride;
                if (hnum >= 0 && wnum >= 0 &&
                    (S1 || (ho * d.stride == hnum && wo * d.stride == wnum)) &&
                    ho < d.Ho && wo < d.Wo)
                    voff = (unsigned)((((a_n[g] * d.Ho + ho) * d.Wo + wo) *
                                       (int64_t)d.K + a_k[g]) * 2);
            }
            if (!SCAT2) {
                int k = a_k[g] + BK;
                while (k >= d.K) {
                    k -= d.K;
                    if (++a_s[g] == d.S) { a_s[g] = 0; ++a_r[g]; }
                }
                a_k[g] = k;
            }
            __builtin_amdgcn_raw_ptr_buffer_load_lds(
                arsrc,
                (__attribute__((address_space(3))) void*)
                    (base + (g * 512 + wid_u * 64) * 8),
                16, voff, 0, 0, 0);
        }
        uint16_t* bbase = base + A_ELEMS;
#pragma unroll
        for (int g = 0; g < NF; ++g) {
            const int64_t off =
                (((int64_t)b_r[g] * d.S + b_s[g]) * d.C + b_col[g]) *
                    (int64_t)d.K + b_k[g];
            int k = b_k[g] + BK;
            while (k >= d.K) {
                k -= d.K;
                if (++b_s[g] == d.S) { b_s[g] = 0; ++b_r[g]; }
            }
            b_k[g] = k;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)
                    (w_rsck + off),
                (__attribute__((address_space(3))) unsigned int*)
                    (bbase + (g * 512 + wid_u * 64) * 8),
                16, 0, 0);
        }
    };

    const int a_row_l = wave_m * 128 + (lane & 15);
    const int frag_kb = (lane >> 4) * 16;
    const int b_col_l = wave_n * (BN / 4) + (lane & 15);

    floatx4 acc[MF][NF] = {};
    int n_stages = rsk / BK;
    if (SPLITK) {
        const int remain = n_stages - blockIdx.z * spz;
        n_stages = remain < spz ? remain : spz;
        if (n_stages <= 0) return;
    }

    auto compute_stage = [&](int buf) {
        const uint16_t* base = lds + buf * BUF_ELEMS;
        const uint16_t* bbase = base + A_ELEMS;
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
            short8 a[MF], b[NF];
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const int row = a_row_l + mf * 16;
                const int byte = (row * 128 + sub * 64 + frag_kb) ^
                                 ((row & 7) << 4);
                a[mf] = *reinterpret_cast<const short8*>(
                    (const char*)base + byte);
            }
#pragma unroll
            for (int nf = 0; nf < NF; ++nf) {
                const int col = b_col_l + nf * 16;
                const int byte = (col * 128 + sub * 64 + frag_kb) ^
                                 ((col & 7) << 4);
                b[nf] = *reinterpret_cast<const short8*>(
                    (const char*)bbase + byte);
            }
            __builtin_amdgcn_s_setprio(1);
            // operands swapped: 4 consecutive C channels per lane in the
            // D-fragment -> one packed 8 B store (see conv_fwd8.hip)
#pragma unroll
            for (int mf = 0; mf < MF; ++mf)
#pragma unroll
                for (int nf = 0; nf < NF; ++nf)
                    acc[mf][nf] = MFMA_BF16(b[nf], a[mf], acc[mf][nf]);
            __builtin_amdgcn_s_setprio(0);
        }
    };

    for (int p = 0; p < DEPTH && p < n_stages; ++p)
        issue_stage(p % BUFS);
    for (int i = 0; i + DEPTH < n_stages; ++i) {
        issue_stage((i + DEPTH) % BUFS);
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(DEPTH * G) : "memory");
        __builtin_amdgcn_s_barrier();
        compute_stage(i % BUFS);
    }
    if (DEPTH == 2 && n_stages >= 2) {   // tail with one stage in flight
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(G) : "memory");
        __builtin_amdgcn_s_barrier();
        compute_stage((n_stages - 2) % BUFS);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    compute_stage((n_stages - 1) % BUFS);

    const int64_t out_row0 = m0 + wave_m * 128 + (lane & 15);
    const int out_col0 = col0 + wave_n * (BN / 4) + (lane >> 4) * 4;
    if (SPLITK) {   // fp32 partials slab [z][M][C]
        float* ws = reinterpret_cast<float*>(dx);
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) {
            const int64_t row = out_row0 + mf * 16;
            if (row < M) {
#pragma unroll
                for (int nf = 0; nf < NF; ++nf)
                    *reinterpret_cast<float4*>(
                        ws + ((int64_t)blockIdx.z * M + row) * d.C +
                        out_col0 + nf * 16) =
                        make_float4(acc[mf][nf][0], acc[mf][nf][1],
                                    acc[mf][nf][2], acc[mf][nf][3]);
            }
        }
        return;
    }
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
        const int64_t row = out_row0 + mf * 16;
        if (row < M) {
            int64_t obase;
            if (SCAT2) {
                const int wo = (int)(row % d.Wo);
                const int ho = (int)((row / d.Wo) % d.Ho);
                const int64_t n = row / ((int64_t)d.Ho * d.Wo);
                obase = ((n * d.H + 2 * ho) * d.W + 2 * wo) * (int64_t)d.C;
            } else {
                obase = row * d.C;
            }
#pragma unroll
            for (int nf = 0; nf < NF; ++nf) {
                ushort4 pk;
#pragma unroll
                for (int rr = 0; rr < 4; ++rr)
                    ((uint16_t*)&pk)[rr] = f32_to_bf16(acc[mf][nf][rr]);
                const int64_t o = obase + out_col0 + nf * 16;
                *reinterpret_cast<ushort4*>(dx + o) = pk;
                if (SCAT2) {   // zero the three odd-position siblings
                    const ushort4 z = {};
                    *reinterpret_cast<ushort4*>(dx + o + d.C) = z;
                    *reinterpret_cast<ushort4*>(dx + o + (int64_t)d.W * d.C) = z;
                    *reinterpret_cast<ushort4*>(
                        dx + o + (int64_t)d.W * d.C + d.C) = z;
                }
            }
        }
    }
}

#include <cstdlib>
extern "C" int conv_dgrad8_plan(ConvDims d, int* bn_out) {
    static int disabled = [] {
        const char* e = getenv("FLASHY_NO_FWD8");
        return e && e[0] == '1';
    }();
    if (disabled) return 0;
    const int64_t M = (int64_t)d.N * d.H * d.W;
    if (d.C % 64 || d.K % 64) return 0;
    const int64_t dout_elems = (int64_t)d.N * d.Ho * d.Wo * d.K;
    if (dout_elems * 2 >= (int64_t)OOB_SENTINEL) return 0;
    const int mtiles = (int)((M + 255) / 256);
    if (d.C % 128 == 0 && (int64_t)mtiles * (d.C / 128) >= 160) {
        *bn_out = 128;
        return mtiles;
    }
    if ((int64_t)mtiles * (d.C / 64) >= 160) {
        *bn_out = 64;
        return mtiles;
    }
    return 0;
}

// 1x1 stride-2 plan: quarter-size GEMM over the output grid (SCAT2)
extern "C" int conv_dgrad8_s2_plan(ConvDims d, int* bn_out) {
    if (d.R != 1 || d.S != 1 || d.stride != 2) return 0;
    if (d.C % 64 || d.K % 64) return 0;
    const int64_t Mq = (int64_t)d.N * d.Ho * d.Wo;
    const int64_t dout_elems = Mq * d.K;
    if (dout_elems * 2 >= (int64_t)OOB_SENTINEL) return 0;
    if (d.W % 2 || d.H % 2) return 0;   // sibling zero-stores assume even
    const int mtiles = (int)((Mq + 255) / 256);
    if (d.C % 128 == 0 && (int64_t)mtiles * (d.C / 128) >= 104) {
        *bn_out = 128;
        return mtiles;
    }
    if ((int64_t)mtiles * (d.C / 64) >= 104) {
        *bn_out = 64;
        return mtiles;
    }
    return 0;
}

extern "C" void launch_conv_dgrad8_s2(const void* dout, const void* w_rsck,
                                      void* dx, ConvDims d, int bn,
                                      int mtiles, hipStream_t stream) {
    dim3 grid((unsigned)mtiles, (unsigned)(d.C / bn));
    const unsigned db =
        (unsigned)((int64_t)d.N * d.Ho * d.Wo * d.K * 2);
    auto dd = (const uint16_t*)dout;
    auto ww = (const uint16_t*)w_rsck;
    auto xx = (uint16_t*)dx;
    if (bn == 128)
        k_conv_dgrad8<128, false, true><<<grid, 512, 0, stream>>>(dd, ww, xx, d, db);
    else
        k_conv_dgrad8<64, false, true><<<grid, 512, 0, stream>>>(dd, ww, xx, d, db);
}

extern "C" void launch_conv_dgrad8(const void* dout, const void* w_rsck,
                                   void* dx, ConvDims d, int bn, int mtiles,
                                   hipStream_t stream) {
    dim3 grid((unsigned)mtiles, (unsigned)(d.C / bn));
    const unsigned db = (unsigned)((int64_t)d.N * d.Ho * d.Wo * d.K * 2);
    auto dd = (const uint16_t*)dout;
    auto ww = (const uint16_t*)w_rsck;
    auto xx = (uint16_t*)dx;
    if (bn == 128) {
        if (d.stride == 1)
            k_conv_dgrad8<128, true><<<grid, 512, 0, stream>>>(dd, ww, xx, d, db);
        else
            k_conv_dgrad8<128, false><<<grid, 512, 0, stream>>>(dd, ww, xx, d, db);
    } else {
        if (d.stride == 1)
            k_conv_dgrad8<64, true><<<grid, 512, 0, stream>>>(dd, ww, xx, d, db);
        else
            k_conv_dgrad8<64, false><<<grid, 512, 0, stream>>>(dd, ww, xx, d, db);
    }
}

extern "C" void launch_conv_dgrad8_splitk(const void* dout,
                                          const void* w_rsck, void* ws,
                                          ConvDims d, int bn, int mtiles,
                                          int spz, int zeff,
                                          hipStream_t stream) {
    dim3 grid((unsigned)mtiles, (unsigned)(d.C / bn), (unsigned)zeff);
    const unsigned db = (unsigned)((int64_t)d.N * d.Ho * d.Wo * d.K * 2);
    auto dd = (const uint16_t*)dout;
    auto ww = (const uint16_t*)w_rsck;
    if (bn == 128) {
        if (d.stride == 1)
            k_conv_dgrad8<128, true, false, true><<<grid, 512, 0, stream>>>(
                dd, ww, (uint16_t*)ws, d, db, spz);
        else
            k_conv_dgrad8<128, false, false, true><<<grid, 512, 0, stream>>>(
                dd, ww, (uint16_t*)ws, d, db, spz);
    } else {
        if (d.stride == 1)
            k_conv_dgrad8<64, true, false, true><<<grid, 512, 0, stream>>>(
                dd, ww, (uint16_t*)ws, d, db, spz);
        else
            k_conv_dgrad8<64, false, false, true><<<grid, 512, 0, stream>>>(
                dd, ww, (uint16_t*)ws, d, db, spz);
    }
}
