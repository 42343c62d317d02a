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

struct ConvDims {
    int N, H, W, C;
    int K, R, S;
    int Ho, Wo;
    int stride, pad;
};

extern "C" {
void launch_conv_fwd(const void* x, const void* w, void* y, ConvDims d,
                     int relu, void* bn_ws, hipStream_t stream);
int conv_fwd_msplit(ConvDims d);
int conv_fwd8_plan(ConvDims d, int* bn_out);
int conv_dgrad8_plan(ConvDims d, int* bn_out);
void launch_conv_fwd8(const void* x, const void* w, void* y, ConvDims d,
                      int relu, void* bn_ws, int bn, int mtiles,
                      hipStream_t stream);
void launch_stem_pad_x(const void* x, void* xp, int64_t N, int H, int W,
                       int Hp, int Wp, int pad, hipStream_t stream);
void launch_stem_pad_w(const void* w, void* wp, int K, int R, int S,
                       hipStream_t stream);
void launch_stem_unpad_dw(const void* dwp, void* dw, int K, int R, int S,
                          hipStream_t stream);
void launch_conv_stem_fwd(const void* x, const void* w, void* y, ConvDims d,
                          hipStream_t stream);
void launch_conv_stem_wgrad(const void* x, const void* dout, void* dw,
                            ConvDims d, hipStream_t stream);
void launch_conv_stem_dgrad(const void* dout, const void* w, void* dx,
                            ConvDims d, hipStream_t stream);
void launch_conv_dgrad(const void* dout, const void* w_rsck, void* dx,
                       ConvDims d, hipStream_t stream);
void launch_conv_fwd_splitk(const void* x, const void* w, void* ws, ConvDims d,
                            int zn, hipStream_t stream);
void launch_conv_dgrad_splitk(const void* dout, const void* w_rsck, void* ws,
                              ConvDims d, int zn, hipStream_t stream);
void launch_splitk_combine(const void* ws, void* out, int64_t total, int zn,
                           int relu, hipStream_t stream);
void launch_weight_transpose(const void* w, void* wt, int K, int rsc,
                             hipStream_t stream);
void launch_weight_transpose_batched(const void* src, void* dst,
                                     const void* meta, int n_convs,
                                     int64_t max_elems, hipStream_t stream);
void launch_conv_wgrad(const void* x, const void* dout, void* dw, ConvDims d,
                       int n_splits, hipStream_t stream);
void launch_bn_stats(const void* x, void* partials, int64_t M, int C,
                     int msplit, hipStream_t stream);
void launch_bn_finalize(const void* partials, int msplit, const void* gamma,
                        const void* beta, void* running_mean,
                        void* running_var, void* work, int64_t M, int C,
                        float eps, float momentum, int update_running,
                        hipStream_t stream);
void launch_bn_apply(const void* x, const void* res, void* y, const void* work,
                     int64_t M, int C, int relu, float slope,
                     hipStream_t stream);
void launch_bn_bwd_reduce(const void* dy, const void* y, const void* x,
                          const void* work, void* dz_out, void* partials,
                          int64_t M, int C, int msplit, int relu, float slope,
                          hipStream_t stream);
void launch_bn_bwd_grads(const void* partials, int msplit, void* bsums,
                         void* dgamma, void* dbeta, int C, hipStream_t stream);
void launch_bn_bwd_apply(const void* dz, const void* x, const void* work,
                         const void* bsums, void* dx, int64_t M, int C,
                         hipStream_t stream);
void launch_fused_sgd(void* p, const void* g, void* m, void* p_bf16, int64_t n,
                      float lr, float momentum, float wd, float grad_scale,
                      int nesterov, hipStream_t stream);
void launch_fused_adam(void* p, const void* g, void* m, void* v, void* p_bf16,
                       void* step_dev, int64_t n, float lr, float beta1,
                       float beta2, float eps, float wd, int64_t step,
                       float grad_scale, int adamw, hipStream_t stream);
void launch_adam_step_inc(void* p, hipStream_t stream);
void launch_maxpool_fwd(const void* x, void* y, void* argmax, ConvDims d,
                        hipStream_t stream);
void launch_maxpool_bwd(const void* dy, const void* argmax, void* dx,
                        ConvDims d, hipStream_t stream);
void launch_cross_entropy(const void* logits, const void* target, void* dlogits,
                          void* loss_sum, int64_t B, int64_t C,
                          float loss_scale, float grad_scale, int is_bf16,
                          hipStream_t stream);
void launch_mse(const void* x, const void* t, void* dx, void* loss_sum,
                int64_t n, float loss_scale, float grad_scale, int is_bf16,
                hipStream_t stream);
void launch_accuracy(const void* logits, const void* target, void* out,
                     int64_t B, int64_t C, int is_bf16, hipStream_t stream);
void launch_linear_fwd(const void* x, const void* w, const void* b, void* y,
                       int64_t B, int64_t I, int64_t O, hipStream_t stream);
void launch_linear_dx(const void* dy, const void* w, void* dx, int64_t B,
                      int64_t I, int64_t O, hipStream_t stream);
void launch_linear_dw(const void* x, const void* dy, void* dw, void* db,
                      int64_t B, int64_t I, int64_t O, hipStream_t stream);
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

static ConvDims make_dims(int N, int H, int W, int C, int K, int R, int S,
                          int Ho, int Wo, int stride, int pad) {
    return ConvDims{N, H, W, C, K, R, S, Ho, Wo, stride, pad};
}

PYBIND11_MODULE(_hip_ops, m) {
    m.doc() = "flashy_amd gfx950 kernels";
    m.attr("ARCH") = "gfx950";

    m.def("conv_fwd",
          [](uintptr_t x, uintptr_t w, uintptr_t y, int N, int H, int W, int C,
             int K, int R, int S, int Ho, int Wo, int stride, int pad,
             bool relu, uintptr_t bn_ws, uintptr_t stream) {
              launch_conv_fwd((const void*)x, (const void*)w, (void*)y,
                              make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                              relu ? 1 : 0, (void*)bn_ws, as_stream(stream));
              check_last();
          });
    m.def("conv_fwd8_direct",
          [](uintptr_t x, uintptr_t w, uintptr_t y, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             bool relu, uintptr_t bn_ws, int bn, int mtiles,
             uintptr_t stream) {
              launch_conv_fwd8((const void*)x, (const void*)w, (void*)y,
                               make_dims(N, H, W, C, K, R, S, Ho, Wo, stride,
                                         pad),
                               relu, (void*)bn_ws, bn, mtiles,
                               as_stream(stream));
              check_last();
          });

    m.def("stem_pad_x",
          [](uintptr_t x, uintptr_t xp, int64_t N, int H, int W, int Hp,
             int Wp, int pad, uintptr_t stream) {
              launch_stem_pad_x((const void*)x, (void*)xp, N, H, W, Hp, Wp,
                                pad, as_stream(stream));
              check_last();
          });

    m.def("stem_pad_w",
          [](uintptr_t w, uintptr_t wp, int K, int R, int S,
             uintptr_t stream) {
              launch_stem_pad_w((const void*)w, (void*)wp, K, R, S,
                                as_stream(stream));
              check_last();
          });

    m.def("stem_unpad_dw",
          [](uintptr_t dwp, uintptr_t dw, int K, int R, int S,
             uintptr_t stream) {
              launch_stem_unpad_dw((const void*)dwp, (void*)dw, K, R, S,
                                   as_stream(stream));
              check_last();
          });

    m.def("conv8_eligible",
          [](int N, int H, int W, int C, int K, int R, int S, int Ho, int Wo,
             int stride, int pad, bool dgrad) {
              int bn = 0;
              ConvDims d = make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad);
              return (dgrad ? conv_dgrad8_plan(d, &bn)
                            : conv_fwd8_plan(d, &bn)) > 0;
          });

    m.def("conv_fwd_msplit",
          [](int N, int H, int W, int C, int K, int R, int S, int Ho, int Wo,
             int stride, int pad) {
              return conv_fwd_msplit(
                  make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad));
          });
    m.def("conv_stem_fwd",
          [](uintptr_t x, uintptr_t w, uintptr_t y, int N, int H, int W, int C,
             int K, int R, int S, int Ho, int Wo, int stride, int pad,
             uintptr_t stream) {
              launch_conv_stem_fwd((const void*)x, (const void*)w, (void*)y,
                                   make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                   as_stream(stream));
              check_last();
          });
    m.def("conv_stem_wgrad",
          [](uintptr_t x, uintptr_t dout, uintptr_t dw, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             uintptr_t stream) {
              launch_conv_stem_wgrad((const void*)x, (const void*)dout,
                                     (void*)dw,
                                     make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                     as_stream(stream));
              check_last();
          });
    m.def("conv_stem_dgrad",
          [](uintptr_t dout, uintptr_t w, uintptr_t dx, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             uintptr_t stream) {
              launch_conv_stem_dgrad((const void*)dout, (const void*)w,
                                     (void*)dx,
                                     make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                     as_stream(stream));
              check_last();
          });
    m.def("conv_dgrad",
          [](uintptr_t dout, uintptr_t w_rsck, uintptr_t dx, int N, int H,
             int W, int C, int K, int R, int S, int Ho, int Wo, int stride,
             int pad, uintptr_t stream) {
              launch_conv_dgrad((const void*)dout, (const void*)w_rsck,
                                (void*)dx,
                                make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                as_stream(stream));
              check_last();
          });
    m.def("conv_fwd_splitk",
          [](uintptr_t x, uintptr_t w, uintptr_t ws, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             int zn, uintptr_t stream) {
              launch_conv_fwd_splitk((const void*)x, (const void*)w, (void*)ws,
                                     make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                     zn, as_stream(stream));
              check_last();
          });
    m.def("conv_dgrad_splitk",
          [](uintptr_t dout, uintptr_t w_rsck, uintptr_t ws, int N, int H,
             int W, int C, int K, int R, int S, int Ho, int Wo, int stride,
             int pad, int zn, uintptr_t stream) {
              launch_conv_dgrad_splitk((const void*)dout, (const void*)w_rsck,
                                       (void*)ws,
                                       make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                       zn, as_stream(stream));
              check_last();
          });
    m.def("splitk_combine",
          [](uintptr_t ws, uintptr_t out, int64_t total, int zn, bool relu,
             uintptr_t stream) {
              launch_splitk_combine((const void*)ws, (void*)out, total, zn,
                                    relu ? 1 : 0, as_stream(stream));
              check_last();
          });
    m.def("weight_transpose",
          [](uintptr_t w, uintptr_t wt, int K, int rsc, uintptr_t stream) {
              launch_weight_transpose((const void*)w, (void*)wt, K, rsc,
                                      as_stream(stream));
              check_last();
          });
    m.def("weight_transpose_batched",
          [](uintptr_t src, uintptr_t dst, uintptr_t meta, int n_convs,
             int64_t max_elems, uintptr_t stream) {
              launch_weight_transpose_batched((const void*)src, (void*)dst,
                                              (const void*)meta, n_convs,
                                              max_elems, as_stream(stream));
              check_last();
          });
    m.def("conv_wgrad",
          [](uintptr_t x, uintptr_t dout, uintptr_t dw, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             int n_splits, uintptr_t stream) {
              launch_conv_wgrad((const void*)x, (const void*)dout, (void*)dw,
                                make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                n_splits, as_stream(stream));
              check_last();
          });
    m.def("bn_stats",
          [](uintptr_t x, uintptr_t partials, int64_t M, int C, int msplit,
             uintptr_t stream) {
              launch_bn_stats((const void*)x, (void*)partials, M, C, msplit,
                              as_stream(stream));
              check_last();
          });
    m.def("bn_finalize",
          [](uintptr_t partials, int msplit, uintptr_t gamma, uintptr_t beta,
             uintptr_t rmean, uintptr_t rvar, uintptr_t work, int64_t M, int C,
             float eps, float momentum, bool update_running, uintptr_t stream) {
              launch_bn_finalize((const void*)partials, msplit,
                                 (const void*)gamma, (const void*)beta,
                                 (void*)rmean, (void*)rvar, (void*)work, M, C,
                                 eps, momentum, update_running ? 1 : 0,
                                 as_stream(stream));
              check_last();
          });
    m.def("bn_apply",
          [](uintptr_t x, uintptr_t res, uintptr_t y, uintptr_t work, int64_t M,
             int C, bool relu, float slope, uintptr_t stream) {
              launch_bn_apply((const void*)x, (const void*)res, (void*)y,
                              (const void*)work, M, C, relu ? 1 : 0, slope,
                              as_stream(stream));
              check_last();
          });
    m.def("bn_bwd_reduce",
          [](uintptr_t dy, uintptr_t y, uintptr_t x, uintptr_t work,
             uintptr_t dz_out, uintptr_t partials, int64_t M, int C,
             int msplit, bool relu, float slope, uintptr_t stream) {
              launch_bn_bwd_reduce((const void*)dy, (const void*)y,
                                   (const void*)x, (const void*)work,
                                   (void*)dz_out, (void*)partials, M, C,
                                   msplit, relu ? 1 : 0, slope,
                                   as_stream(stream));
              check_last();
          });
    m.def("bn_bwd_grads",
          [](uintptr_t partials, int msplit, uintptr_t bsums, uintptr_t dgamma,
             uintptr_t dbeta, int C, uintptr_t stream) {
              launch_bn_bwd_grads((const void*)partials, msplit, (void*)bsums,
                                  (void*)dgamma, (void*)dbeta, C,
                                  as_stream(stream));
              check_last();
          });
    m.def("bn_bwd_apply",
          [](uintptr_t dz, uintptr_t x, uintptr_t work, uintptr_t bsums,
             uintptr_t dx, int64_t M, int C, uintptr_t stream) {
              launch_bn_bwd_apply((const void*)dz, (const void*)x,
                                  (const void*)work, (const void*)bsums,
                                  (void*)dx, M, C, as_stream(stream));
              check_last();
          });

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
             uintptr_t p_bf16, uintptr_t step_dev, int64_t n, float lr,
             float beta1, float beta2, float eps, float wd, int64_t step,
             float grad_scale, bool adamw, uintptr_t stream) {
              launch_fused_adam((void*)p, (const void*)g, (void*)mom,
                                (void*)var, (void*)p_bf16, (void*)step_dev, n,
                                lr, beta1, beta2, eps, wd, step, grad_scale,
                                adamw ? 1 : 0, as_stream(stream));
              check_last();
          });
    m.def("adam_step_inc", [](uintptr_t p, uintptr_t stream) {
        launch_adam_step_inc((void*)p, as_stream(stream));
        check_last();
    });
    m.def("maxpool_fwd",
          [](uintptr_t x, uintptr_t y, uintptr_t argmax, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             uintptr_t stream) {
              launch_maxpool_fwd((const void*)x, (void*)y, (void*)argmax,
                                 make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
                                 as_stream(stream));
              check_last();
          });
    m.def("maxpool_bwd",
          [](uintptr_t dy, uintptr_t argmax, uintptr_t dx, int N, int H, int W,
             int C, int K, int R, int S, int Ho, int Wo, int stride, int pad,
             uintptr_t stream) {
              launch_maxpool_bwd((const void*)dy, (const void*)argmax,
                                 (void*)dx,
                                 make_dims(N, H, W, C, K, R, S, Ho, Wo, stride, pad),
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

    m.def("mse",
          [](uintptr_t x, uintptr_t t, uintptr_t dx, uintptr_t loss_sum,
             int64_t n, float loss_scale, float grad_scale, bool is_bf16,
             uintptr_t stream) {
              launch_mse((const void*)x, (const void*)t, (void*)dx,
                         (void*)loss_sum, n, loss_scale, grad_scale,
                         is_bf16 ? 1 : 0, as_stream(stream));
              check_last();
          });

    m.def("accuracy",
          [](uintptr_t logits, uintptr_t target, uintptr_t out, int64_t B,
             int64_t C, bool is_bf16, uintptr_t stream) {
              launch_accuracy((const void*)logits, (const void*)target,
                              (void*)out, B, C, is_bf16 ? 1 : 0,
                              as_stream(stream));
              check_last();
          });

    m.def("linear_fwd",
          [](uintptr_t x, uintptr_t w, uintptr_t b, uintptr_t y, int64_t B,
             int64_t I, int64_t O, uintptr_t stream) {
              launch_linear_fwd((const void*)x, (const void*)w,
                                (const void*)b, (void*)y, B, I, O,
                                as_stream(stream));
              check_last();
          });

    m.def("linear_dx",
          [](uintptr_t dy, uintptr_t w, uintptr_t dx, int64_t B, int64_t I,
             int64_t O, uintptr_t stream) {
              launch_linear_dx((const void*)dy, (const void*)w, (void*)dx, B,
                               I, O, as_stream(stream));
              check_last();
          });

    m.def("linear_dw",
          [](uintptr_t x, uintptr_t dy, uintptr_t dw, uintptr_t db, int64_t B,
             int64_t I, int64_t O, uintptr_t stream) {
              launch_linear_dw((const void*)x, (const void*)dy, (void*)dw,
                               (void*)db, B, I, O, as_stream(stream));
              check_last();
          });
}
