// Copyright (c) Flashy-AMD authors.
// pybind11 bindings for the gfx950 kernels.  Torch-ABI-free by design: the
// Python wrappers (flashy_amd/ops/__init__.py) pass raw device pointers
// (tensor.data_ptr()) and the current HIP stream handle
// (torch.cuda.current_stream().cuda_stream); all launches go onto that
// stream, so they are captured by HIP graphs like any other kernel.

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>

#include <cstdint>
#include <stdexcept>

namespace py = pybind11;

extern "C" {
void launch_fused_sgd(void* p, const void* g, void* m, void* p_bf16, int64_t n,
                      float lr, float momentum, float wd, float grad_scale,
                      int nesterov, hipStream_t stream);
void launch_fused_adam(void* p, const void* g, void* m, void* v, void* p_bf16,
                       int64_t n, float lr, float beta1, float beta2, float eps,
                       float wd, int64_t step, float grad_scale, int adamw,
                       hipStream_t stream);
void launch_cross_entropy(const void* logits, const void* target, void* dlogits,
                          void* loss_sum, int64_t B, int64_t C,
                          float loss_scale, float grad_scale, int is_bf16,
                          hipStream_t stream);
void launch_bce_logits(const void* x, void* dx, void* loss_sum, int64_t n,
                       float target, float loss_scale, float grad_scale,
                       int is_bf16, hipStream_t stream);
}

static void check_last() {
    hipError_t err = hipGetLastError();
    if (err != hipSuccess)
        throw std::runtime_error(std::string("HIP launch failed: ") +
                                 hipGetErrorString(err));
}

static hipStream_t as_stream(uintptr_t s) { return reinterpret_cast<hipStream_t>(s); }

PYBIND11_MODULE(_hip_ops, m) {
    m.doc() = "flashy_amd gfx950 kernels";
    m.attr("ARCH") = "gfx950";

    m.def("fused_sgd",
          [](uintptr_t p, uintptr_t g, uintptr_t mom, uintptr_t p_bf16,
             int64_t n, float lr, float momentum, float wd, float grad_scale,
             bool nesterov, uintptr_t stream) {
              launch_fused_sgd((void*)p, (const void*)g, (void*)mom,
                               (void*)p_bf16, n, lr, momentum, wd, grad_scale,
                               nesterov ? 1 : 0, as_stream(stream));
              check_last();
          });

    m.def("fused_adam",
          [](uintptr_t p, uintptr_t g, uintptr_t mom, uintptr_t var,
             uintptr_t p_bf16, int64_t n, float lr, float beta1, float beta2,
             float eps, float wd, int64_t step, float grad_scale, bool adamw,
             uintptr_t stream) {
              launch_fused_adam((void*)p, (const void*)g, (void*)mom,
                                (void*)var, (void*)p_bf16, n, lr, beta1, beta2,
                                eps, wd, step, grad_scale, adamw ? 1 : 0,
                                as_stream(stream));
              check_last();
          });

    m.def("cross_entropy",
          [](uintptr_t logits, uintptr_t target, uintptr_t dlogits,
             uintptr_t loss_sum, int64_t B, int64_t C, float loss_scale,
             float grad_scale, bool is_bf16, uintptr_t stream) {
              launch_cross_entropy((const void*)logits, (const void*)target,
                                   (void*)dlogits, (void*)loss_sum, B, C,
                                   loss_scale, grad_scale, is_bf16 ? 1 : 0,
                                   as_stream(stream));
              check_last();
          });

    m.def("bce_logits",
          [](uintptr_t x, uintptr_t dx, uintptr_t loss_sum, int64_t n,
             float target, float loss_scale, float grad_scale, bool is_bf16,
             uintptr_t stream) {
              launch_bce_logits((const void*)x, (void*)dx, (void*)loss_sum, n,
                                target, loss_scale, grad_scale,
                                is_bf16 ? 1 : 0, as_stream(stream));
              check_last();
          });
}
