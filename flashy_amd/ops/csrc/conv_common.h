// Copyright (c) Flashy-AMD authors.
// Shared definitions for the NHWC bf16 implicit-GEMM convolution kernels
// (gfx950 MFMA 16x16x32 bf16).
#pragma once

#include "common.h"

// All tensors NHWC (= torch channels_last memory format), bf16 payloads as
// uint16.  Weights: fwd consumes torch's channels_last conv weight layout
// [K][R][S][C] ("KRSC"); dgrad consumes a pre-transposed copy [R][S][C][K]
// ("RSCK", built per-step by k_weight_transpose).

struct ConvDims {
    int N, H, W, C;     // input
    int K, R, S;        // filters
    int Ho, Wo;         // output spatial
    int stride, pad;
};

// MFMA fragment helpers for v_mfma_f32_16x16x32_bf16 (per the CDNA4 guide §3):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7 (8 bf16)
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   D: lane l holds D[row = (l>>4)*4 + r][col = l&15], r = 0..3 (4 f32)
// (verified on hardware by tests/test_conv_gpu.py with asymmetric operands)

#define MFMA_BF16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

// Block geometry shared by conv fwd and dgrad kernels.
#define CONV_BM 128      // GEMM rows (output pixels) per block
#define CONV_BN 64       // GEMM cols (channels) per block
#define CONV_BK 32       // reduction chunk (one MFMA K)
#define CONV_THREADS 256 // 4 waves: 2 (m) x 2 (n)
// LDS A-tile pitch in bf16 elements: 48 keeps every 16-lane ds_read_b128
// group on distinct banks (see design notes) and 16B alignment.
#define CONV_APITCH 40  // 80 B rows: 16 B-aligned b128, stride 20 dwords (16 distinct
                        // banks mod 64), and the BM128 double buffer fits 4
                        // blocks/CU (49 KB at pitch 48 capped it at 3)
