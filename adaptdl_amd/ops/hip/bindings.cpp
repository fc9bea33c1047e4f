// Python bindings for the adaptdl_amd CDNA4 kernels (gns_kernels.hip).
// Launches on the current torch HIP stream so ordering with autograd-
// produced gradients and RCCL all-reduces is inherited from stream order.
// Kernel launches live in gns_kernels.hip (hipcc-compiled); this file is
// host-only glue.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <cmath>

extern "C" {
void launch_sqsum(const float*, long, double*, hipStream_t);
void launch_scale_sqsum(float*, long, float, double*, hipStream_t);
void launch_sqsum_diff_update(const float*, float*, long, double*,
                              hipStream_t);
void launch_sqsum_avg(const float*, const float*, long, double*,
                      hipStream_t);
void launch_precond_sqsum(const float*, const float*, long, float, float,
                          double*, hipStream_t);
void launch_fused_sgd(float*, const float*, float*, long, float, float,
                      float, float, int, int, hipStream_t);
void launch_fused_adamw(float*, const float*, float*, float*, long, float,
                        float, float, float, float, float, float, int,
                        hipStream_t);
}

namespace {

void check_f32(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
    TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be float32");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

void check_out(const torch::Tensor& out) {
    TORCH_CHECK(out.is_cuda(), "out must be a GPU tensor");
    TORCH_CHECK(out.scalar_type() == torch::kFloat64, "out must be float64");
}

hipStream_t stream() {
    return at::hip::getCurrentHIPStream().stream();
}

void sqsum(torch::Tensor x, torch::Tensor out) {
    check_f32(x, "x"); check_out(out);
    long n = x.numel();
    if (n == 0) return;
    launch_sqsum(x.data_ptr<float>(), n, out.data_ptr<double>(), stream());
}

void scale_and_sqsum(torch::Tensor x, double scale, torch::Tensor out) {
    check_f32(x, "x"); check_out(out);
    long n = x.numel();
    if (n == 0) return;
    launch_scale_sqsum(x.data_ptr<float>(), n, (float)scale,
                       out.data_ptr<double>(), stream());
}

void sqsum_diff_update(torch::Tensor cur, torch::Tensor prev,
                       torch::Tensor out) {
    check_f32(cur, "cur"); check_f32(prev, "prev"); check_out(out);
    TORCH_CHECK(cur.numel() == prev.numel(), "cur/prev size mismatch");
    long n = cur.numel();
    if (n == 0) return;
    launch_sqsum_diff_update(cur.data_ptr<float>(), prev.data_ptr<float>(),
                             n, out.data_ptr<double>(), stream());
}

void sqsum_avg(torch::Tensor cur, torch::Tensor prev, torch::Tensor out) {
    check_f32(cur, "cur"); check_f32(prev, "prev"); check_out(out);
    TORCH_CHECK(cur.numel() == prev.numel(), "cur/prev size mismatch");
    long n = cur.numel();
    if (n == 0) return;
    launch_sqsum_avg(cur.data_ptr<float>(), prev.data_ptr<float>(), n,
                     out.data_ptr<double>(), stream());
}

void precond_sqsum(torch::Tensor g, torch::Tensor v, double beta2,
                   double eps, long step, torch::Tensor out) {
    check_f32(g, "g"); check_f32(v, "v"); check_out(out);
    TORCH_CHECK(g.numel() == v.numel(), "g/v size mismatch");
    long n = g.numel();
    if (n == 0) return;
    double corr = 1.0 - std::pow(beta2, (double)step);
    float inv_corr_sqrt = (float)(1.0 / std::sqrt(corr));
    launch_precond_sqsum(g.data_ptr<float>(), v.data_ptr<float>(), n,
                         inv_corr_sqrt, (float)eps, out.data_ptr<double>(),
                         stream());
}

void fused_sgd(torch::Tensor p, torch::Tensor g, torch::Tensor m, double lr,
               double momentum, double weight_decay, double dampening,
               bool nesterov) {
    check_f32(p, "p"); check_f32(g, "g");
    long n = p.numel();
    if (n == 0) return;
    bool has_m = momentum != 0.0;
    if (has_m) {
        check_f32(m, "momentum_buf");
        TORCH_CHECK(m.numel() == n, "momentum size mismatch");
    }
    launch_fused_sgd(p.data_ptr<float>(), g.data_ptr<float>(),
                     has_m ? m.data_ptr<float>() : nullptr, n, (float)lr,
                     (float)momentum, (float)weight_decay, (float)dampening,
                     (int)nesterov, (int)has_m, stream());
}

void fused_adamw(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                 torch::Tensor v, double lr, double beta1, double beta2,
                 double eps, double weight_decay, long step,
                 bool adam_mode) {
    check_f32(p, "p"); check_f32(g, "g"); check_f32(m, "m"); check_f32(v, "v");
    long n = p.numel();
    if (n == 0) return;
    float bias1 = (float)(1.0 - std::pow(beta1, (double)step));
    float bias2 = (float)(1.0 - std::pow(beta2, (double)step));
    launch_fused_adamw(p.data_ptr<float>(), g.data_ptr<float>(),
                       m.data_ptr<float>(), v.data_ptr<float>(), n,
                       (float)lr, (float)beta1, (float)beta2, (float)eps,
                       (float)weight_decay, bias1, bias2, (int)adam_mode,
                       stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
    mod.def("sqsum", &sqsum, "out += sum(x^2) in fp64");
    mod.def("scale_and_sqsum", &scale_and_sqsum,
            "x *= scale; out += sum(x^2) fused");
    mod.def("sqsum_diff_update", &sqsum_diff_update,
            "out += sum((cur-prev)^2); prev = cur");
    mod.def("sqsum_avg", &sqsum_avg, "out += sum(((cur+prev)/2)^2)");
    mod.def("precond_sqsum", &precond_sqsum,
            "out += sum((g/pinv)^2), Adam preconditioner");
    mod.def("fused_sgd", &fused_sgd, "fused flat-bucket SGD step");
    mod.def("fused_adamw", &fused_adamw, "fused flat-bucket Adam(W) step");
}
