// Copyright (c) Flashy-AMD authors.
// Halo-dedup implicit-GEMM conv for gfx950 — stride-1 same-size 3x3-class
// layers (C in {64,128}).
//
// The fwd8 kernel re-stages the A (im2col) operand once per 64-deep stage:
// for a 3x3 conv that is a 9x amplification of the input bytes through
// L2/L3 (PMC: those shapes are A-staging-traffic bound).  Here each block
// covers 256 consecutive output pixels WITHIN one image and DMAs the
// compact input slab — the spanned rows plus halo, with one zero pixel
// column on each side — into LDS ONCE; the whole rsc reduction then reads
// MFMA fragments straight out of the slab with per-lane tap addressing.
// Only the small weight tile still cycles through LDS buffers per stage.
//
//   slab[(hv - row_lo) * (W + 2*pad) + (wo + s)][c]  <->  x[n][hv][wi][c]
//   (hv rows outside the image and the side columns load hardware zeros
//    via the buffer-bounds OOB sentinel, so tap reads need no branches)
//
// Same conventions as conv_fwd8.hip: 512 threads (2M x 4N waves), BN in
// {64,128} over K, BK=64, source-side XOR swizzle + matching read XOR,
// swapped MFMA operands -> packed 8 B stores, fused BN-stats epilogue.
// The dgrad of these layers is dispatched through the same kernel with
// remapped dims (A slab = dy, B = the RSCK-transposed weight, taps
// reflected — see launch_conv_dedup_dgrad).

#include "conv_common.h"

#define OOB_SENTINEL 0xF0000000u

template <int BN, bool RELU>
__global__ void __launch_bounds__(512, 2)
k_conv_dedup(const uint16_t* __restrict__ x, const uint16_t* __restrict__ w,
             uint16_t* __restrict__ y, float* __restrict__ bn_ws,
             ConvDims d, unsigned x_nbytes, int tpi, int slab_rows,
             int lds_elems) {
    constexpr int NF = BN / 64;
    constexpr int MF = 8;
    constexpr int B_ELEMS = BN * 64;
    constexpr int BBUF = 3;

    const int rsc = d.R * d.S * d.C;
    const int64_t hw = (int64_t)d.Ho * d.Wo;
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wid_u = __builtin_amdgcn_readfirstlane(tid >> 6);
    const int wave_m = wid_u >> 2;
    const int wave_n = wid_u & 3;

    unsigned bx = blockIdx.x;
    if ((gridDim.x & 7) == 0)
        bx = (bx & 7) * (gridDim.x >> 3) + (bx >> 3);
    const int64_t n_img = bx / tpi;
    const int tile = (int)(bx % tpi);
    const int64_t pix0 = (int64_t)tile * 256;   // within this image
    const int col0 = blockIdx.y * BN;

    const int Wp = d.W + 2 * d.pad;             // slab row width in pixels
    const int ho0 = (int)(pix0 / d.Wo);
    const int row_lo = ho0 - d.pad;             // first slab row (virtual)

    extern __shared__ __attribute__((aligned(16))) uint16_t lds[];  // [slab | B ring]
    uint16_t* const bring = lds + (lds_elems - BBUF * B_ELEMS);

    const auto xrsrc = __builtin_amdgcn_make_buffer_rsrc(
        (void*)x, 0, x_nbytes, 0x00020000);

    // ---- slab DMA: every 16 B chunk computes its own source ------------
    const int cpp = d.C / 8;                    // chunks per pixel
    const int slab_pixels = slab_rows * Wp;
    const int slab_chunks = slab_pixels * cpp;
    for (int chunk = tid; chunk < slab_chunks; chunk += 512) {
        const int pix = chunk / cpp;
        const int kc_d = chunk - pix * cpp;
        const int kc_s = kc_d ^ (pix & 7);      // source-side swizzle
        const int hv = row_lo + pix / Wp;
        const int wi = pix % Wp - d.pad;
        unsigned voff = OOB_SENTINEL;
        if (hv >= 0 && hv < d.H && wi >= 0 && wi < d.W)
            voff = (unsigned)((((n_img * d.H + hv) * d.W + wi) *
                               (int64_t)d.C + kc_s * 8) * 2);
        __builtin_amdgcn_raw_ptr_buffer_load_lds(
            xrsrc,
            (__attribute__((address_space(3))) void*)(lds + chunk * 8),
            16, voff, 0, 0, 0);
    }

    // ---- B staging (per 64-deep stage, 3-buffer ring) -------------------
    unsigned b_src[2];
#pragma unroll
    for (int g = 0; g < NF; ++g) {
        const int chunk = g * 512 + tid;
        const int col = chunk >> 3;
        const int kc_s = (chunk & 7) ^ (col & 7);
        b_src[g] = (unsigned)(((int64_t)(col0 + col) * rsc + kc_s * 8) * 2);
    }
    auto issue_b = [&](int stage) {
        uint16_t* base = bring + (stage % BBUF) * B_ELEMS;
        const unsigned kb = (unsigned)(stage * 64 * 2);
#pragma unroll
        for (int g = 0; g < NF; ++g)
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)
                    ((const char*)w + b_src[g] + kb),
                (__attribute__((address_space(3))) unsigned int*)
                    (base + (g * 512 + wid_u * 64) * 8),
                16, 0, 0);
    };

    // ---- per-lane A tap state ------------------------------------------
    // lane reads pixel p = pix0 + wave_m*128 + mf*16 + (lane&15) at tap
    // slot (lane>>4)*8 within each 32-deep sub; the (r, s, c) walk advances
    // +64 per stage.  Slab address: ((ho-ho0+r)*Wp + wo + s)*C + c.
    const int64_t p_l = pix0 + wave_m * 128 + (lane & 15);   // + mf*16
    int a_hobase[MF];   // (ho - ho0) * Wp + wo  per m fragment
    bool a_ok[MF];
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
        const int64_t p = p_l + mf * 16;
        a_ok[mf] = p < hw;
        const int64_t pc = a_ok[mf] ? p : hw - 1;
        a_hobase[mf] = (int)((pc / d.Wo - ho0) * Wp + pc % d.Wo);
    }
    int t_r, t_s, t_c;
    {
        const int kk = (lane >> 4) * 8;
        t_r = kk / (d.S * d.C);
        const int sc = kk - t_r * d.S * d.C;
        t_s = sc / d.C;
        t_c = sc - t_s * d.C;
    }
    const int b_col_l = wave_n * (BN / 4) + (lane & 15);
    const int frag_kb = (lane >> 4) * 16;

    floatx4 acc[MF][NF] = {};
    const int n_stages = rsc / 64;

    issue_b(0);
    if (n_stages > 1) issue_b(1);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // slab + B0 + B1 landed
    __builtin_amdgcn_s_barrier();

    for (int i = 0; i < n_stages; ++i) {
        if (i + 1 < n_stages)
            asm volatile("s_waitcnt vmcnt(%0)" ::"i"(NF) : "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();   // all waves done with slot (i+2)%3
        if (i + 2 < n_stages) issue_b(i + 2);
        const uint16_t* bbase = bring + (i % BBUF) * B_ELEMS;
#pragma unroll
        for (int sub = 0; sub < 2; ++sub) {
            // tap for this lane this sub: walk state holds sub 0; sub 1 is
            // +32 taps — recompute cheaply from the walk with a carry
            int r = t_r, s = t_s, c = t_c + sub * 32;
            while (c >= d.C) {
                c -= d.C;
                if (++s == d.S) { s = 0; ++r; }
            }
            const int sp_off = r * Wp + s;   // slab-pixel offset of the tap
            short8 a[MF], b[NF];
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const int slabpix = a_hobase[mf] + sp_off;
                const int byte = ((slabpix * d.C + c) * 2) ^
                                 ((slabpix & 7) << 4);
                a[mf] = *reinterpret_cast<const short8*>(
                    (const char*)lds + byte);
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
#pragma unroll
            for (int mf = 0; mf < MF; ++mf)
#pragma unroll
                for (int nf = 0; nf < NF; ++nf)
                    acc[mf][nf] = MFMA_BF16(b[nf], a[mf], acc[mf][nf]);
            __builtin_amdgcn_s_setprio(0);
        }
        // advance the walk one stage (+64)
        {
            int c = t_c + 64;
            while (c >= d.C) {
                c -= d.C;
                if (++t_s == d.S) { t_s = 0; ++t_r; }
            }
            t_c = c;
        }
    }

    // ---- epilogue (same layout as conv_fwd8) ---------------------------
    const int out_col0 = col0 + wave_n * (BN / 4) + (lane >> 4) * 4;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
        if (a_ok[mf]) {
            const int64_t row = n_img * hw + p_l + mf * 16;
#pragma unroll
            for (int nf = 0; nf < NF; ++nf) {
                ushort4 pk;
#pragma unroll
                for (int rr = 0; rr < 4; ++rr) {
                    float v = acc[mf][nf][rr];
                    if (RELU) v = fmaxf(v, 0.f);
                    ((uint16_t*)&pk)[rr] = f32_to_bf16(v);
                }
                *reinterpret_cast<ushort4*>(
                    y + row * d.K + out_col0 + nf * 16) = pk;
            }
        }
    }

    if (bn_ws != nullptr) {
        __syncthreads();
        float* sred = reinterpret_cast<float*>(lds);
#pragma unroll
        for (int nf = 0; nf < NF; ++nf) {
            float s[4] = {}, s2[4] = {};
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
                for (int rr = 0; rr < 4; ++rr) {
                    float v = acc[mf][nf][rr];
                    if (RELU) v = fmaxf(v, 0.f);
                    if (!a_ok[mf]) v = 0.f;
                    s[rr] += v;
                    s2[rr] = fmaf(v, v, s2[rr]);
                }
            }
#pragma unroll
            for (int off = 1; off < 16; off <<= 1)
#pragma unroll
                for (int rr = 0; rr < 4; ++rr) {
                    s[rr] += __shfl_xor(s[rr], off, 64);
                    s2[rr] += __shfl_xor(s2[rr], off, 64);
                }
            if ((lane & 15) == 0) {
                const int colL = wave_n * (BN / 4) + nf * 16 + (lane >> 4) * 4;
                *reinterpret_cast<float4*>(&sred[wave_m * BN + colL]) =
                    make_float4(s[0], s[1], s[2], s[3]);
                *reinterpret_cast<float4*>(&sred[(2 + wave_m) * BN + colL]) =
                    make_float4(s2[0], s2[1], s2[2], s2[3]);
            }
        }
        __syncthreads();
        if (tid < BN) {
            const float s = sred[tid] + sred[BN + tid];
            const float s2 = sred[2 * BN + tid] + sred[3 * BN + tid];
            const int c = col0 + tid;
            bn_ws[(int64_t)c * gridDim.x + blockIdx.x] = s;
            bn_ws[((int64_t)d.K + c) * gridDim.x + blockIdx.x] = s2;
        }
    }
}

// Eligibility: stride-1 same-size conv, C in {64,128}, K % 64 == 0, slab +
// B ring fit in LDS.  Returns grid.x (N * tiles-per-image) or 0; outputs
// bn, tiles-per-image, slab rows and the dynamic-LDS element count.
extern "C" int conv_dedup_plan(ConvDims d, int* bn_out, int* tpi_out,
                               int* rows_out, int* elems_out) {
    static int disabled = [] {
        const char* e = getenv("FLASHY_NO_FWD8");
        return e && e[0] == '1';
    }();
    if (disabled) return 0;
    if (d.stride != 1 || d.Ho != d.H || d.Wo != d.W) return 0;
    if (d.R != 2 * d.pad + 1 || d.S != 2 * d.pad + 1 || d.R < 2) return 0;
    if ((d.C != 64 && d.C != 128) || d.K % 64) return 0;
    const int64_t x_elems = (int64_t)d.N * d.H * d.W * d.C;
    if (x_elems * 2 >= (int64_t)OOB_SENTINEL) return 0;
    const int64_t hw = (int64_t)d.Ho * d.Wo;
    const int tpi = (int)((hw + 255) / 256);
    // worst-case spanned output rows of a 256-pixel tile + halo
    const int rows = (int)((255 / d.Wo) + 1 + 2 * d.pad + 1);
    const int Wp = d.W + 2 * d.pad;
    const int bn = d.K % 128 == 0 ? 128 : 64;
    const int slab_elems = rows * Wp * d.C;
    const int lds_elems = slab_elems + 3 * bn * 64;
    if (lds_elems * 2 > 160 * 1024) return 0;
    const int64_t grid = (int64_t)d.N * tpi * (d.K / bn);
    if (grid < 160) return 0;
    *bn_out = bn;
    *tpi_out = tpi;
    *rows_out = rows;
    *elems_out = lds_elems;
    return (int)((int64_t)d.N * tpi);
}

extern "C" void launch_conv_dedup(const void* x, const void* w, void* y,
                                  ConvDims d, int relu, void* bn_ws, int bn,
                                  int tpi, int rows, int lds_elems,
                                  hipStream_t stream) {
    dim3 grid((unsigned)((int64_t)d.N * tpi), (unsigned)(d.K / bn));
    const unsigned xb = (unsigned)((int64_t)d.N * d.H * d.W * d.C * 2);
    const size_t shmem = (size_t)lds_elems * 2;
    static bool attr_set = [] {
        const int cap = 160 * 1024;
        (void)hipFuncSetAttribute(
            (const void*)&k_conv_dedup<128, true>,
            hipFuncAttributeMaxDynamicSharedMemorySize, cap);
        (void)hipFuncSetAttribute(
            (const void*)&k_conv_dedup<128, false>,
            hipFuncAttributeMaxDynamicSharedMemorySize, cap);
        (void)hipFuncSetAttribute(
            (const void*)&k_conv_dedup<64, true>,
            hipFuncAttributeMaxDynamicSharedMemorySize, cap);
        (void)hipFuncSetAttribute(
            (const void*)&k_conv_dedup<64, false>,
            hipFuncAttributeMaxDynamicSharedMemorySize, cap);
        return true;
    }();
    (void)attr_set;
    auto xx = (const uint16_t*)x;
    auto ww = (const uint16_t*)w;
    auto yy = (uint16_t*)y;
    if (bn == 128) {
        if (relu)
            k_conv_dedup<128, true><<<grid, 512, shmem, stream>>>(
                xx, ww, yy, (float*)bn_ws, d, xb, tpi, rows, lds_elems);
        else
            k_conv_dedup<128, false><<<grid, 512, shmem, stream>>>(
                xx, ww, yy, (float*)bn_ws, d, xb, tpi, rows, lds_elems);
    } else {
        if (relu)
            k_conv_dedup<64, true><<<grid, 512, shmem, stream>>>(
                xx, ww, yy, (float*)bn_ws, d, xb, tpi, rows, lds_elems);
        else
            k_conv_dedup<64, false><<<grid, 512, shmem, stream>>>(
                xx, ww, yy, (float*)bn_ws, d, xb, tpi, rows, lds_elems);
    }
}
