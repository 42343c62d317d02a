// Copyright (c) Flashy-AMD authors.
// Fused flat-buffer optimizers for gfx950.
//
// The framework's FusedSGD/FusedAdam optimizers keep ALL parameters (and
// their gradients / momenta) as views into single contiguous fp32 buffers
// (flashy_amd/optim.py), so one kernel launch updates the whole model:
// perfectly coalesced float4 traffic, grid-stride over <=2048 blocks — the
// memory-bound roofline shape for HBM3E (guidelines G11/G13 of the CDNA4
// guide).  Replaces the per-parameter optimizer loops the reference inherits
// from torch (reference hot-op inventory: SURVEY.md §2.10 SGD/Adam rows).

#include "common.h"

#include <math.h>

// ---------------------------------------------------------------------------
// SGD (torch.optim.SGD semantics):
//   d = g + wd * p
//   if momentum: m = mu * m + d ; d = nesterov ? d + mu * m : m
//   p -= lr * d
// ---------------------------------------------------------------------------

struct SgdArgs {
    float* __restrict__ p;
    const float* __restrict__ g;
    float* __restrict__ m;          // momentum buffer (null = no momentum)
    uint16_t* __restrict__ p_bf16;  // optional bf16 mirror of p (may be null)
    int64_t n;
    float lr, momentum, wd, grad_scale;
    int nesterov;
};

__device__ __forceinline__ float sgd_one(const SgdArgs& a, float p, float g,
                                         float& m) {
    g = g * a.grad_scale + a.wd * p;
    if (a.momentum != 0.f) {
        m = m * a.momentum + g;
        g = a.nesterov ? g + a.momentum * m : m;
    }
    return p - a.lr * g;
}

template <bool HAS_M, bool HAS_BF16>
__global__ void __launch_bounds__(256)
k_fused_sgd(SgdArgs a) {
    const int64_t n4 = a.n / 4;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    float4* p4 = reinterpret_cast<float4*>(a.p);
    const float4* g4 = reinterpret_cast<const float4*>(a.g);
    float4* m4 = reinterpret_cast<float4*>(a.m);
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        float4 p = p4[i];
        float4 g = g4[i];
        float4 m = HAS_M ? m4[i] : float4{0, 0, 0, 0};
        p.x = sgd_one(a, p.x, g.x, m.x);
        p.y = sgd_one(a, p.y, g.y, m.y);
        p.z = sgd_one(a, p.z, g.z, m.z);
        p.w = sgd_one(a, p.w, g.w, m.w);
        p4[i] = p;
        if (HAS_M) m4[i] = m;
        if (HAS_BF16) {
            ushort4 b;
            b.x = f32_to_bf16(p.x); b.y = f32_to_bf16(p.y);
            b.z = f32_to_bf16(p.z); b.w = f32_to_bf16(p.w);
            reinterpret_cast<ushort4*>(a.p_bf16)[i] = b;
        }
    }
    // scalar tail
    for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < a.n;
         i += stride) {
        float m = HAS_M ? a.m[i] : 0.f;
        float p = sgd_one(a, a.p[i], a.g[i], m);
        a.p[i] = p;
        if (HAS_M) a.m[i] = m;
        if (HAS_BF16) a.p_bf16[i] = f32_to_bf16(p);
    }
}

extern "C" void launch_fused_sgd(void* p, const void* g, void* m, void* p_bf16,
                                 int64_t n, float lr, float momentum, float wd,
                                 float grad_scale, int nesterov,
                                 hipStream_t stream) {
    SgdArgs a{(float*)p, (const float*)g, (float*)m, (uint16_t*)p_bf16,
              n, lr, momentum, wd, grad_scale, nesterov};
    int grid = ew_grid(n / 4 + 1, 256, 1);
    if (m != nullptr && p_bf16 != nullptr)
        k_fused_sgd<true, true><<<grid, 256, 0, stream>>>(a);
    else if (m != nullptr)
        k_fused_sgd<true, false><<<grid, 256, 0, stream>>>(a);
    else if (p_bf16 != nullptr)
        k_fused_sgd<false, true><<<grid, 256, 0, stream>>>(a);
    else
        k_fused_sgd<false, false><<<grid, 256, 0, stream>>>(a);
}

// ---------------------------------------------------------------------------
// Adam / AdamW (torch.optim semantics with bias correction):
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
//   p -= lr * (m/bc1) / (sqrt(v/bc2) + eps)   [+ decoupled or L2 wd]
// ---------------------------------------------------------------------------

struct AdamArgs {
    float* __restrict__ p;
    const float* __restrict__ g;
    float* __restrict__ m;
    float* __restrict__ v;
    uint16_t* __restrict__ p_bf16;
    const long long* step_ptr;  // device step (graph-safe); null -> use bc1/bc2
    int64_t n;
    float lr, beta1, beta2, eps, wd, bc1, bc2, grad_scale;
    int adamw;  // 1: decoupled weight decay, 0: L2 into grad
};

__device__ __forceinline__ float adam_one(const AdamArgs& a, float p, float g,
                                          float& m, float& v) {
    g *= a.grad_scale;
    if (!a.adamw) g += a.wd * p;
    else p *= (1.f - a.lr * a.wd);
    m = a.beta1 * m + (1.f - a.beta1) * g;
    v = a.beta2 * v + (1.f - a.beta2) * g * g;
    return p - a.lr * (m / a.bc1) / (sqrtf(v / a.bc2) + a.eps);
}

template <bool HAS_BF16>
__global__ void __launch_bounds__(256)
k_fused_adam(AdamArgs a) {
    if (a.step_ptr != nullptr) {
        const float t = (float)*a.step_ptr;
        a.bc1 = 1.f - powf(a.beta1, t);
        a.bc2 = 1.f - powf(a.beta2, t);
    }
    const int64_t n4 = a.n / 4;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    float4* p4 = reinterpret_cast<float4*>(a.p);
    const float4* g4 = reinterpret_cast<const float4*>(a.g);
    float4* m4 = reinterpret_cast<float4*>(a.m);
    float4* v4 = reinterpret_cast<float4*>(a.v);
    for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4; i += stride) {
        float4 p = p4[i];
        float4 g = g4[i];
        float4 m = m4[i];
        float4 v = v4[i];
        p.x = adam_one(a, p.x, g.x, m.x, v.x);
        p.y = adam_one(a, p.y, g.y, m.y, v.y);
        p.z = adam_one(a, p.z, g.z, m.z, v.z);
        p.w = adam_one(a, p.w, g.w, m.w, v.w);
        p4[i] = p;
        m4[i] = m;
        v4[i] = v;
        if (HAS_BF16) {
            ushort4 b;
            b.x = f32_to_bf16(p.x); b.y = f32_to_bf16(p.y);
            b.z = f32_to_bf16(p.z); b.w = f32_to_bf16(p.w);
            reinterpret_cast<ushort4*>(a.p_bf16)[i] = b;
        }
    }
    for (int64_t i = n4 * 4 + blockIdx.x * blockDim.x + threadIdx.x; i < a.n;
         i += stride) {
        float m = a.m[i], v = a.v[i];
        float p = adam_one(a, a.p[i], a.g[i], m, v);
        a.p[i] = p;
        a.m[i] = m;
        a.v[i] = v;
        if (HAS_BF16) a.p_bf16[i] = f32_to_bf16(p);
    }
}

__global__ void k_step_inc(long long* p) {
    if (threadIdx.x == 0 && blockIdx.x == 0) ++(*p);
}

extern "C" void launch_adam_step_inc(void* p, hipStream_t stream) {
    k_step_inc<<<1, 1, 0, stream>>>((long long*)p);
}

extern "C" void launch_fused_adam(void* p, const void* g, void* m, void* v,
                                  void* p_bf16, void* step_dev, int64_t n,
                                  float lr, float beta1, float beta2, float eps,
                                  float wd, int64_t step, float grad_scale,
                                  int adamw, hipStream_t stream) {
    AdamArgs a{(float*)p, (const float*)g, (float*)m, (float*)v,
               (uint16_t*)p_bf16, (const long long*)step_dev, n, lr, beta1,
               beta2, eps, wd,
               1.f - powf(beta1, (float)step), 1.f - powf(beta2, (float)step),
               grad_scale, adamw};
    int grid = ew_grid(n / 4 + 1, 256, 1);
    if (p_bf16 != nullptr)
        k_fused_adam<true><<<grid, 256, 0, stream>>>(a);
    else
        k_fused_adam<false><<<grid, 256, 0, stream>>>(a);
}
