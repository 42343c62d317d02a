// Copyright (c) Flashy-AMD authors.
// Fused loss kernels for gfx950: cross-entropy and BCE-with-logits.
//
// Training always needs the backward, so the forward kernel also produces
// the input gradient in the same pass (one read of the logits instead of
// three across separate softmax/nll/backward kernels).  The scalar loss is
// accumulated with one float atomicAdd per wave (guideline G12).
// Replaces the reference workloads' F.cross_entropy /
// F.binary_cross_entropy_with_logits call sites.

#include "common.h"

#include <math.h>

// ---------------------------------------------------------------------------
// Cross-entropy over logits [B, C] with int64 targets [B].
// One wave per row (C up to a few thousand; lanes stride the row).
// Outputs: dlogits [B, C] = (softmax - onehot) * grad_scale, and
// loss_sum[0] += sum_b (lse_b - logit_b[target_b]) * loss_scale.
// T: 0 = float32 logits, 1 = bfloat16 logits (fp32 math inside).
// ---------------------------------------------------------------------------

template <typename T>
__device__ __forceinline__ float load_f32(const T* p, int64_t i);
template <>
__device__ __forceinline__ float load_f32<float>(const float* p, int64_t i) {
    return p[i];
}
template <>
__device__ __forceinline__ float load_f32<uint16_t>(const uint16_t* p, int64_t i) {
    return bf16_to_f32(p[i]);
}

template <typename T>
__device__ __forceinline__ void store_f32(T* p, int64_t i, float v);
template <>
__device__ __forceinline__ void store_f32<float>(float* p, int64_t i, float v) {
    p[i] = v;
}
template <>
__device__ __forceinline__ void store_f32<uint16_t>(uint16_t* p, int64_t i, float v) {
    p[i] = f32_to_bf16(v);
}

template <typename T>
__global__ void __launch_bounds__(256)
k_cross_entropy(const T* __restrict__ logits, const int64_t* __restrict__ target,
                T* __restrict__ dlogits, float* __restrict__ loss_sum,
                int64_t B, int64_t C, float loss_scale, float grad_scale) {
    const int wave = threadIdx.x / WAVE_SIZE;          // 4 waves per block
    const int lane = threadIdx.x % WAVE_SIZE;
    const int64_t row = (int64_t)blockIdx.x * 4 + wave;
    if (row >= B) return;
    const T* x = logits + row * C;
    T* dx = dlogits + row * C;
    const int64_t tgt = target[row];

    float mx = -INFINITY;
    for (int64_t c = lane; c < C; c += WAVE_SIZE)
        mx = fmaxf(mx, load_f32(x, c));
    mx = wave_bcast(wave_max(mx));

    float sum = 0.f;
    for (int64_t c = lane; c < C; c += WAVE_SIZE)
        sum += expf(load_f32(x, c) - mx);
    sum = wave_bcast(wave_sum(sum));
    const float inv_sum = 1.f / sum;
    const float lse = mx + logf(sum);

    for (int64_t c = lane; c < C; c += WAVE_SIZE) {
        float p = expf(load_f32(x, c) - mx) * inv_sum;
        float grad = (p - (c == tgt ? 1.f : 0.f)) * grad_scale;
        store_f32(dx, c, grad);
    }
    if (lane == 0) {
        float loss = (lse - load_f32(x, tgt)) * loss_scale;
        atomicAdd(loss_sum, loss);
    }
}

extern "C" void launch_cross_entropy(const void* logits, const void* target,
                                     void* dlogits, void* loss_sum, int64_t B,
                                     int64_t C, float loss_scale,
                                     float grad_scale, int is_bf16,
                                     hipStream_t stream) {
    int grid = (int)((B + 3) / 4);
    if (is_bf16)
        k_cross_entropy<uint16_t><<<grid, 256, 0, stream>>>(
            (const uint16_t*)logits, (const int64_t*)target, (uint16_t*)dlogits,
            (float*)loss_sum, B, C, loss_scale, grad_scale);
    else
        k_cross_entropy<float><<<grid, 256, 0, stream>>>(
            (const float*)logits, (const int64_t*)target, (float*)dlogits,
            (float*)loss_sum, B, C, loss_scale, grad_scale);
}

// ---------------------------------------------------------------------------
// BCE-with-logits against a CONSTANT target (the GAN convention:
// flashy_amd/adversarial.py trains towards D(fake)=1 / D(real)=0):
//   loss_i = max(x,0) - x*t + log1p(exp(-|x|))
//   dx_i   = (sigmoid(x) - t) * grad_scale
// Elementwise + one wave-level atomic for the mean.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(256)
k_bce_logits(const T* __restrict__ x, T* __restrict__ dx,
             float* __restrict__ loss_sum, int64_t n, float target,
             float loss_scale, float grad_scale) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    float local = 0.f;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        float v = load_f32(x, i);
        local += fmaxf(v, 0.f) - v * target + log1pf(expf(-fabsf(v)));
        float sig = 1.f / (1.f + expf(-v));
        store_f32(dx, i, (sig - target) * grad_scale);
    }
    local = wave_sum(local);
    if ((threadIdx.x % WAVE_SIZE) == 0 && local != 0.f)
        atomicAdd(loss_sum, local * loss_scale);
}

extern "C" void launch_bce_logits(const void* x, void* dx, void* loss_sum,
                                  int64_t n, float target, float loss_scale,
                                  float grad_scale, int is_bf16,
                                  hipStream_t stream) {
    int grid = ew_grid(n, 256, 4);
    if (is_bf16)
        k_bce_logits<uint16_t><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, (uint16_t*)dx, (float*)loss_sum, n, target,
            loss_scale, grad_scale);
    else
        k_bce_logits<float><<<grid, 256, 0, stream>>>(
            (const float*)x, (float*)dx, (float*)loss_sum, n, target,
            loss_scale, grad_scale);
}

// ---------------------------------------------------------------------------
// MSE loss, fused fwd+grad:  loss += (x-t)^2 * loss_scale,
// dx = 2(x-t) * grad_scale.  Elementwise + one atomic per wave.
// Replaces F.mse_loss in the teacher-student workload
// (/root/reference/tests/dummy/train.py:93).
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(256)
k_mse(const T* __restrict__ x, const T* __restrict__ t, T* __restrict__ dx,
      float* __restrict__ loss_sum, int64_t n, float loss_scale,
      float grad_scale) {
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    float local = 0.f;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += stride) {
        const float d = load_f32(x, i) - load_f32(t, i);
        local = fmaf(d, d, local);
        store_f32(dx, i, 2.f * d * grad_scale);
    }
    local = wave_sum(local);
    if ((threadIdx.x % WAVE_SIZE) == 0 && local != 0.f)
        atomicAdd(loss_sum, local * loss_scale);
}

extern "C" void launch_mse(const void* x, const void* t, void* dx,
                           void* loss_sum, int64_t n, float loss_scale,
                           float grad_scale, int is_bf16, hipStream_t stream) {
    int grid = ew_grid(n, 256, 4);
    if (is_bf16)
        k_mse<uint16_t><<<grid, 256, 0, stream>>>(
            (const uint16_t*)x, (const uint16_t*)t, (uint16_t*)dx,
            (float*)loss_sum, n, loss_scale, grad_scale);
    else
        k_mse<float><<<grid, 256, 0, stream>>>(
            (const float*)x, (const float*)t, (float*)dx, (float*)loss_sum, n,
            loss_scale, grad_scale);
}

// ---------------------------------------------------------------------------
// Accuracy: mean(argmax(logits, 1) == target).  One wave per row, atomic
// count of correct rows into out[0] (caller pre-zeroes and divides by B).
// Replaces the torch argmax+eq+mean chain (3 launches -> 1,
// examples/cifar/solver.py accuracy metric).
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(256)
k_accuracy(const T* __restrict__ logits, const int64_t* __restrict__ target,
           float* __restrict__ out, int64_t B, int64_t C) {
    const int wave = threadIdx.x / WAVE_SIZE;
    const int lane = threadIdx.x % WAVE_SIZE;
    const int64_t row = (int64_t)blockIdx.x * 4 + wave;
    if (row >= B) return;
    const T* x = logits + row * C;
    float best = -INFINITY;
    int64_t best_c = 0;
    for (int64_t c = lane; c < C; c += WAVE_SIZE) {
        const float v = load_f32(x, c);
        if (v > best || (v == best && c < best_c)) { best = v; best_c = c; }
    }
    // wave-reduce (value, index), ties -> smallest index (torch argmax)
    for (int off = WAVE_SIZE / 2; off; off >>= 1) {
        const float ov = __shfl_down(best, off, WAVE_SIZE);
        const int64_t oc = __shfl_down(best_c, off, WAVE_SIZE);
        if (ov > best || (ov == best && oc < best_c)) { best = ov; best_c = oc; }
    }
    if (lane == 0 && best_c == target[row]) atomicAdd(out, 1.f);
}

extern "C" void launch_accuracy(const void* logits, const void* target,
                                void* out, int64_t B, int64_t C, int is_bf16,
                                hipStream_t stream) {
    int grid = (int)((B + 3) / 4);
    if (is_bf16)
        k_accuracy<uint16_t><<<grid, 256, 0, stream>>>(
            (const uint16_t*)logits, (const int64_t*)target, (float*)out, B, C);
    else
        k_accuracy<float><<<grid, 256, 0, stream>>>(
            (const float*)logits, (const int64_t*)target, (float*)out, B, C);
}
