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
void launch_precond_sqsum_dev(const float*, const float*, long,
                              const float*, double*, hipStream_t);
void launch_sqsum_bf16(const unsigned short*, long, double*, hipStream_t);
void launch_scale_sqsum_bf16(unsigned short*, long, float, double*,
                             hipStream_t);
void launch_sqsum_diff_update_bf16(const unsigned short*, unsigned short*,
                                   long, double*, hipStream_t);
void launch_sqsum_avg_bf16(const unsigned short*, const unsigned short*,
                           long, double*, hipStream_t);
void launch_sqsum_fp16(const unsigned short*, long, double*, hipStream_t);
void launch_scale_sqsum_fp16(unsigned short*, long, float, double*,
                             hipStream_t);
void launch_sqsum_diff_update_fp16(const unsigned short*, unsigned short*,
                                   long, double*, hipStream_t);
void launch_sqsum_avg_fp16(const unsigned short*, const unsigned short*,
                           long, double*, hipStream_t);
void launch_fused_sgd(float*, const float*, float*, long, float, float,
                      float, float, int, int, hipStream_t);
void launch_fused_adamw(float*, const float*, float*, float*, long, float,
                        float, float, float, float, float, float, int,
                        hipStream_t);
void launch_bn_fwd(const unsigned short*, const unsigned short*,
                   unsigned short*, long, int, const float*, const float*,
                   float*, float*, float, float, int, int, float*, float*,
                   float*, float*, float*, float*, hipStream_t);
void launch_bn_bwd(const unsigned short*, const unsigned short*,
                   const unsigned short*, unsigned short*, unsigned short*,
                   long, int, const float*, const float*, const float*,
                   const float*, const float*, int, int, float*, float*,
                   float*, float*, float*, hipStream_t);
int conv3x3_wrw_supported(int, int, int, int);
int conv3x3_mm_supported(int, int, int, int);
int conv3x3_s2_bwd_supported(int, int, int, int);
int conv3x3_s2_fwd_supported(int, int, int, int);
int conv3x3_s2_wrw_supported(int, int, int, int);
int conv3x3_s2_bwd_w8b_supported(int, int, int, int);
void launch_conv3x3_s2_bwd_w8b(const unsigned short*,
                               const unsigned short*, unsigned short*,
                               int, int, int, int, hipStream_t);
int conv3x3_s2_wrw_nsplit(int, int, int, int, int);
void launch_conv3x3_s2_wrw(const unsigned short*, const unsigned short*,
                           float*, float*, int, int, int, int, int,
                           hipStream_t);
void launch_conv3x3_s2_fwd(const unsigned short*, const unsigned short*,
                           unsigned short*, int, int, int, int, int,
                           hipStream_t);
void launch_conv3x3_s2_bwd(const unsigned short*, const unsigned short*,
                           unsigned short*, int, int, int, int, int,
                           hipStream_t);
void launch_conv3x3_mm(const unsigned short*, const unsigned short*,
                       unsigned short*, int, int, int, int, int,
                       hipStream_t);
int conv3x3_wrw_nsplit(int, int, int, int, int);
void launch_conv3x3_wrw(const unsigned short*, const unsigned short*,
                        float*, float*, int, int, int, int, int,
                        hipStream_t);
}

// Mirror of the kernel-side stage-1 grid sizing (bn_kernels.hip).
static long bn_num_rblocks(long m, long c) {
    long rpb = 256 / (c / 8);
    if (rpb < 1) rpb = 1;
    long nb = (m + rpb - 1) / rpb;
    if (nb > 1024) nb = 1024;
    if (nb < 1) nb = 1;
    return nb;
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

// The statistics entry points accept float32 (master-grad training)
// plus bfloat16/float16 (true-low-precision-parameter models) buckets.
bool is_bf16(const torch::Tensor& t) {
    return t.scalar_type() == torch::kBFloat16;
}

bool is_fp16(const torch::Tensor& t) {
    return t.scalar_type() == torch::kHalf;
}

void check_stat_in(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
    TORCH_CHECK(t.scalar_type() == torch::kFloat32 ||
                t.scalar_type() == torch::kBFloat16 ||
                t.scalar_type() == torch::kHalf,
                name, " must be float32, bfloat16, or float16");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

unsigned short* bf16_ptr(torch::Tensor& t) {
    return reinterpret_cast<unsigned short*>(t.data_ptr<at::BFloat16>());
}

unsigned short* fp16_ptr(torch::Tensor& t) {
    return reinterpret_cast<unsigned short*>(t.data_ptr<at::Half>());
}

void sqsum(torch::Tensor x, torch::Tensor out) {
    check_stat_in(x, "x"); check_out(out);
    long n = x.numel();
    if (n == 0) return;
    if (is_bf16(x))
        launch_sqsum_bf16(bf16_ptr(x), n, out.data_ptr<double>(),
                          stream());
    else if (is_fp16(x))
        launch_sqsum_fp16(fp16_ptr(x), n, out.data_ptr<double>(),
                          stream());
    else
        launch_sqsum(x.data_ptr<float>(), n, out.data_ptr<double>(),
                     stream());
}

void scale_and_sqsum(torch::Tensor x, double scale, torch::Tensor out) {
    check_stat_in(x, "x"); check_out(out);
    long n = x.numel();
    if (n == 0) return;
    if (is_bf16(x))
        launch_scale_sqsum_bf16(bf16_ptr(x), n, (float)scale,
                                out.data_ptr<double>(), stream());
    else if (is_fp16(x))
        launch_scale_sqsum_fp16(fp16_ptr(x), n, (float)scale,
                                out.data_ptr<double>(), stream());
    else
        launch_scale_sqsum(x.data_ptr<float>(), n, (float)scale,
                           out.data_ptr<double>(), stream());
}

void sqsum_diff_update(torch::Tensor cur, torch::Tensor prev,
                       torch::Tensor out) {
    check_stat_in(cur, "cur"); check_stat_in(prev, "prev");
    check_out(out);
    TORCH_CHECK(cur.numel() == prev.numel(), "cur/prev size mismatch");
    TORCH_CHECK(cur.scalar_type() == prev.scalar_type(),
                "cur/prev dtype mismatch");
    long n = cur.numel();
    if (n == 0) return;
    if (is_bf16(cur))
        launch_sqsum_diff_update_bf16(bf16_ptr(cur), bf16_ptr(prev), n,
                                      out.data_ptr<double>(), stream());
    else if (is_fp16(cur))
        launch_sqsum_diff_update_fp16(fp16_ptr(cur), fp16_ptr(prev), n,
                                      out.data_ptr<double>(), stream());
    else
        launch_sqsum_diff_update(cur.data_ptr<float>(),
                                 prev.data_ptr<float>(), n,
                                 out.data_ptr<double>(), stream());
}

void sqsum_avg(torch::Tensor cur, torch::Tensor prev, torch::Tensor out) {
    check_stat_in(cur, "cur"); check_stat_in(prev, "prev");
    check_out(out);
    TORCH_CHECK(cur.numel() == prev.numel(), "cur/prev size mismatch");
    TORCH_CHECK(cur.scalar_type() == prev.scalar_type(),
                "cur/prev dtype mismatch");
    long n = cur.numel();
    if (n == 0) return;
    if (is_bf16(cur))
        launch_sqsum_avg_bf16(bf16_ptr(cur), bf16_ptr(prev), n,
                              out.data_ptr<double>(), stream());
    else if (is_fp16(cur))
        launch_sqsum_avg_fp16(fp16_ptr(cur), fp16_ptr(prev), n,
                              out.data_ptr<double>(), stream());
    else
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

void precond_sqsum_dev(torch::Tensor g, torch::Tensor v, torch::Tensor pc,
                       torch::Tensor out) {
    // hipGraph-capturable variant: bias-correction scalars pc =
    // {inv_corr_sqrt, eps, use_precond, unused} live in device memory
    // (updated by the eager fused-Adam step between replays).
    check_f32(g, "g"); check_f32(v, "v"); check_out(out);
    check_f32(pc, "pc");
    TORCH_CHECK(g.numel() == v.numel(), "g/v size mismatch");
    TORCH_CHECK(pc.numel() >= 3, "pc must hold >= 3 scalars");
    long n = g.numel();
    if (n == 0) return;
    launch_precond_sqsum_dev(g.data_ptr<float>(), v.data_ptr<float>(), n,
                             pc.data_ptr<float>(), out.data_ptr<double>(),
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

// ---- fused NHWC BatchNorm(+ReLU) -------------------------------------

void check_bn_x(const torch::Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
    TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name,
                " must be bfloat16");
    TORCH_CHECK(t.dim() == 4, name, " must be 4-D NCHW-shaped");
    TORCH_CHECK(t.is_contiguous(at::MemoryFormat::ChannelsLast), name,
                " must be channels_last");
}

void check_vecf(const torch::Tensor& t, long n, const char* name) {
    TORCH_CHECK(t.is_cuda() && t.scalar_type() == torch::kFloat32 &&
                t.is_contiguous() && t.numel() == n,
                name, " must be a contiguous fp32 GPU tensor of size ", n);
}

// y = [relu](bn(x)); fills sums(2C ws), save_mean/save_rstd/scale/shift(C).
void bn_fwd(torch::Tensor x, torch::Tensor z, torch::Tensor y,
            torch::Tensor gamma, torch::Tensor beta,
            torch::Tensor running_mean, torch::Tensor running_var,
            double momentum, double eps, bool train, bool relu,
            torch::Tensor ws, torch::Tensor sums,
            torch::Tensor save_mean, torch::Tensor save_rstd,
            torch::Tensor scale, torch::Tensor shift) {
    check_bn_x(x, "x"); check_bn_x(y, "y");
    const unsigned short* zp = nullptr;
    if (z.defined() && z.numel() > 0) {
        check_bn_x(z, "z");
        TORCH_CHECK(z.sizes() == x.sizes(), "residual shape mismatch");
        zp = (const unsigned short*)z.data_ptr();
    }
    const long c = x.size(1);
    const long m = x.numel() / c;
    TORCH_CHECK(c % 8 == 0 && c / 8 <= 256,
                "bn_fwd requires C % 8 == 0 and C <= 2048, got C=", c);
    check_vecf(gamma, c, "gamma"); check_vecf(beta, c, "beta");
    if (train)
        check_vecf(ws, 2 * c * bn_num_rblocks(m, c), "ws");
    check_vecf(sums, 2 * c, "sums");
    check_vecf(save_mean, c, "save_mean");
    check_vecf(save_rstd, c, "save_rstd");
    check_vecf(scale, c, "scale"); check_vecf(shift, c, "shift");
    float* rm = nullptr;
    float* rv = nullptr;
    if (running_mean.defined() && running_mean.numel() > 0) {
        check_vecf(running_mean, c, "running_mean");
        check_vecf(running_var, c, "running_var");
        rm = running_mean.data_ptr<float>();
        rv = running_var.data_ptr<float>();
    } else {
        TORCH_CHECK(train, "eval-mode bn_fwd requires running statistics");
    }
    launch_bn_fwd((const unsigned short*)x.data_ptr(), zp,
                  (unsigned short*)y.data_ptr(), m, (int)c,
                  gamma.data_ptr<float>(), beta.data_ptr<float>(), rm, rv,
                  (float)momentum, (float)eps, (int)train, (int)relu,
                  train ? ws.data_ptr<float>() : nullptr,
                  sums.data_ptr<float>(), save_mean.data_ptr<float>(),
                  save_rstd.data_ptr<float>(), scale.data_ptr<float>(),
                  shift.data_ptr<float>(), stream());
}

void bn_bwd(torch::Tensor x, torch::Tensor z, torch::Tensor dy,
            torch::Tensor dx, torch::Tensor dz,
            torch::Tensor gamma, torch::Tensor save_mean,
            torch::Tensor save_rstd, torch::Tensor scale,
            torch::Tensor shift, bool train, bool relu, torch::Tensor ws,
            torch::Tensor sums, torch::Tensor dgamma, torch::Tensor dbeta,
            torch::Tensor pqr) {
    check_bn_x(x, "x"); check_bn_x(dy, "dy"); check_bn_x(dx, "dx");
    const unsigned short* zp = nullptr;
    unsigned short* dzp = nullptr;
    if (z.defined() && z.numel() > 0) {
        check_bn_x(z, "z"); check_bn_x(dz, "dz");
        TORCH_CHECK(z.sizes() == x.sizes(), "residual shape mismatch");
        zp = (const unsigned short*)z.data_ptr();
        dzp = (unsigned short*)dz.data_ptr();
    }
    const long c = x.size(1);
    const long m = x.numel() / c;
    check_vecf(gamma, c, "gamma");
    check_vecf(save_mean, c, "save_mean");
    check_vecf(save_rstd, c, "save_rstd");
    check_vecf(scale, c, "scale"); check_vecf(shift, c, "shift");
    check_vecf(ws, 2 * c * bn_num_rblocks(m, c), "ws");
    check_vecf(sums, 2 * c, "sums");
    check_vecf(dgamma, c, "dgamma"); check_vecf(dbeta, c, "dbeta");
    check_vecf(pqr, 3 * c, "pqr");
    launch_bn_bwd((const unsigned short*)x.data_ptr(), zp,
                  (const unsigned short*)dy.data_ptr(),
                  (unsigned short*)dx.data_ptr(), dzp, m, (int)c,
                  gamma.data_ptr<float>(), save_mean.data_ptr<float>(),
                  save_rstd.data_ptr<float>(), scale.data_ptr<float>(),
                  shift.data_ptr<float>(), (int)train, (int)relu,
                  ws.data_ptr<float>(), sums.data_ptr<float>(),
                  dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                  pqr.data_ptr<float>(), stream());
}

// ---- custom MFMA conv 3x3 wrw ---------------------------------------

bool conv_wrw_ok(long N, long H, long W, long C, long K) {
    return conv3x3_wrw_supported((int)H, (int)W, (int)C, (int)K) != 0;
}

long conv_wrw_nsplit(long N, long H, long W, long C, long K) {
    return conv3x3_wrw_nsplit((int)N, (int)H, (int)W, (int)C, (int)K);
}

// dw = wrw(x, dy): x (N,C,H,W) channels_last bf16, dy (N,K,H,W)
// channels_last bf16, dw (K,C,3,3) channels_last fp32, ws workspace.
void conv_wrw(torch::Tensor x, torch::Tensor dy, torch::Tensor ws,
              torch::Tensor dw) {
    check_bn_x(x, "x"); check_bn_x(dy, "dy");
    const long N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
    const long K = dy.size(1);
    TORCH_CHECK(dy.size(0) == N && dy.size(2) == H && dy.size(3) == W,
                "dy shape mismatch");
    TORCH_CHECK(conv_wrw_ok(N, H, W, C, K), "unsupported wrw shape");
    TORCH_CHECK(dw.is_cuda() && dw.scalar_type() == torch::kFloat32 &&
                dw.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                dw.size(0) == K && dw.size(1) == C && dw.size(2) == 3 &&
                dw.size(3) == 3, "dw must be (K,C,3,3) channels_last f32");
    const long nsplit = conv_wrw_nsplit(N, H, W, C, K);
    check_vecf(ws, nsplit * K * 9 * C, "ws");
    launch_conv3x3_wrw((const unsigned short*)x.data_ptr(),
                       (const unsigned short*)dy.data_ptr(),
                       ws.data_ptr<float>(), dw.data_ptr<float>(),
                       (int)N, (int)H, (int)W, (int)C, (int)K, stream());
}

bool conv_mm_ok(long N, long H, long W, long C, long K) {
    return conv3x3_mm_supported((int)H, (int)W, (int)C, (int)K) != 0;
}

// y = conv3x3_s1p1(x, w): x (N,C,H,W) channels_last bf16; w EITHER the
// (K,C,3,3) channels_last weight (forward) or the host-prepared
// (C,K,3,3)-channels_last flipped transform (backward-data with C and K
// swapped roles by the caller).
void conv_mm(torch::Tensor x, torch::Tensor w, torch::Tensor y) {
    check_bn_x(x, "x"); check_bn_x(y, "y");
    const long N = x.size(0), C = x.size(1), H = x.size(2),
        Wd = x.size(3);
    const long K = y.size(1);
    TORCH_CHECK(y.size(0) == N && y.size(2) == H && y.size(3) == Wd,
                "y shape mismatch");
    TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
                w.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                w.size(0) == K && w.size(1) == C && w.size(2) == 3 &&
                w.size(3) == 3, "w must be (K,C,3,3) channels_last bf16");
    TORCH_CHECK(conv_mm_ok(N, H, Wd, C, K), "unsupported conv_mm shape");
    launch_conv3x3_mm((const unsigned short*)x.data_ptr(),
                      (const unsigned short*)w.data_ptr(),
                      (unsigned short*)y.data_ptr(),
                      (int)N, (int)H, (int)Wd, (int)C, (int)K, stream());
}

bool conv_s2_bwd_ok(long N, long Ho, long Wo, long K, long C) {
    (void)N;
    return conv3x3_s2_bwd_supported((int)Ho, (int)Wo, (int)K, (int)C)
        != 0;
}

// dx = conv3x3_s2p1_backward_data(dy, w): dy (N,K,Ho,Wo) channels_last
// bf16; wt is the host-prepared (C,3,3,K) CONTIGUOUS transposed weight
// (w.permute(1,2,3,0).contiguous()); dx (N,C,2Ho,2Wo) channels_last.
void conv_s2_bwd(torch::Tensor dy, torch::Tensor wt, torch::Tensor dx) {
    check_bn_x(dy, "dy"); check_bn_x(dx, "dx");
    const long N = dy.size(0), K = dy.size(1), Ho = dy.size(2),
        Wo = dy.size(3);
    const long C = dx.size(1);
    TORCH_CHECK(dx.size(0) == N && dx.size(2) == 2 * Ho &&
                dx.size(3) == 2 * Wo, "dx shape mismatch");
    TORCH_CHECK(wt.is_cuda() && wt.scalar_type() == torch::kBFloat16 &&
                wt.is_contiguous() && wt.size(0) == C &&
                wt.size(1) == 3 && wt.size(2) == 3 && wt.size(3) == K,
                "wt must be (C,3,3,K) contiguous bf16");
    TORCH_CHECK(conv_s2_bwd_ok(N, Ho, Wo, K, C),
                "unsupported s2 bwd-data shape");
    launch_conv3x3_s2_bwd((const unsigned short*)dy.data_ptr(),
                          (const unsigned short*)wt.data_ptr(),
                          (unsigned short*)dx.data_ptr(),
                          (int)N, (int)Ho, (int)Wo, (int)K, (int)C,
                          stream());
}

bool conv_s2_fwd_ok(long N, long H, long W, long C, long K) {
    (void)N;
    return conv3x3_s2_fwd_supported((int)H, (int)W, (int)C, (int)K) != 0;
}

// y = conv3x3_s2p1(x, w): x (N,C,H,W) channels_last bf16, w (K,C,3,3)
// channels_last bf16, y (N,K,H/2,W/2) channels_last (experimental --
// production forward stays on MIOpen pending a per-shape win).
void conv_s2_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor y) {
    check_bn_x(x, "x"); check_bn_x(y, "y");
    const long N = x.size(0), C = x.size(1), H = x.size(2),
        Wd = x.size(3);
    const long K = y.size(1);
    TORCH_CHECK(y.size(0) == N && y.size(2) == H / 2 &&
                y.size(3) == Wd / 2, "y shape mismatch");
    TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 &&
                w.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                w.size(0) == K && w.size(1) == C && w.size(2) == 3 &&
                w.size(3) == 3, "w must be (K,C,3,3) channels_last bf16");
    TORCH_CHECK(conv_s2_fwd_ok(N, H, Wd, C, K),
                "unsupported s2 fwd shape");
    launch_conv3x3_s2_fwd((const unsigned short*)x.data_ptr(),
                          (const unsigned short*)w.data_ptr(),
                          (unsigned short*)y.data_ptr(),
                          (int)N, (int)H, (int)Wd, (int)C, (int)K,
                          stream());
}

bool conv_s2_wrw_ok(long N, long Ho, long Wo, long C, long K) {
    (void)N;
    return conv3x3_s2_wrw_supported((int)Ho, (int)Wo, (int)C, (int)K)
        != 0;
}

long conv_s2_wrw_nsplit(long N, long Ho, long Wo, long C, long K) {
    return conv3x3_s2_wrw_nsplit((int)N, (int)Ho, (int)Wo, (int)C,
                                 (int)K);
}

// dW = wrw(x, dy) for the stride-2 3x3 conv: x (N,C,2Ho,2Wo)
// channels_last bf16, dy (N,K,Ho,Wo) channels_last bf16, ws fp32
// workspace (nsplit*K*9*C), dw (K,C,3,3) channels_last fp32 output.
void conv_s2_wrw(torch::Tensor x, torch::Tensor dy, torch::Tensor ws,
                 torch::Tensor dw) {
    check_bn_x(x, "x"); check_bn_x(dy, "dy");
    const long N = dy.size(0), K = dy.size(1), Ho = dy.size(2),
        Wo = dy.size(3);
    const long C = x.size(1);
    TORCH_CHECK(x.size(0) == N && x.size(2) == 2 * Ho &&
                x.size(3) == 2 * Wo, "x shape mismatch");
    TORCH_CHECK(conv_s2_wrw_ok(N, Ho, Wo, C, K),
                "unsupported s2 wrw shape");
    const long nsplit = conv_s2_wrw_nsplit(N, Ho, Wo, C, K);
    TORCH_CHECK(ws.is_cuda() && ws.scalar_type() == torch::kFloat32 &&
                ws.numel() >= nsplit * K * 9 * C, "ws too small");
    TORCH_CHECK(dw.is_cuda() && dw.scalar_type() == torch::kFloat32 &&
                dw.is_contiguous(at::MemoryFormat::ChannelsLast) &&
                dw.size(0) == K && dw.size(1) == C, "dw must be "
                "(K,C,3,3) channels_last fp32");
    launch_conv3x3_s2_wrw((const unsigned short*)x.data_ptr(),
                          (const unsigned short*)dy.data_ptr(),
                          ws.data_ptr<float>(), dw.data_ptr<float>(),
                          (int)N, (int)Ho, (int)Wo, (int)C, (int)K,
                          stream());
}

bool conv_s2_bwd_w8b_ok(long N, long Ho, long Wo, long K, long C) {
    (void)N;
    return conv3x3_s2_bwd_w8b_supported((int)Ho, (int)Wo, (int)K,
                                        (int)C) != 0;
}

// Wo=8 stride-2 bwd-data, P2=4 redesign (see conv_kernels.hip).
void conv_s2_bwd_w8b(torch::Tensor dy, torch::Tensor wt,
                     torch::Tensor dx) {
    check_bn_x(dy, "dy"); check_bn_x(dx, "dx");
    const long N = dy.size(0), K = dy.size(1), Ho = dy.size(2),
        Wo = dy.size(3);
    const long C = dx.size(1);
    TORCH_CHECK(dx.size(0) == N && dx.size(2) == 2 * Ho &&
                dx.size(3) == 2 * Wo, "dx shape mismatch");
    TORCH_CHECK(wt.is_cuda() && wt.scalar_type() == torch::kBFloat16 &&
                wt.is_contiguous() && wt.size(0) == C &&
                wt.size(3) == K, "wt must be (C,3,3,K) contiguous bf16");
    TORCH_CHECK(conv_s2_bwd_w8b_ok(N, Ho, Wo, K, C),
                "unsupported w8b shape");
    launch_conv3x3_s2_bwd_w8b((const unsigned short*)dy.data_ptr(),
                              (const unsigned short*)wt.data_ptr(),
                              (unsigned short*)dx.data_ptr(),
                              (int)N, (int)Ho, (int)K, (int)C,
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
    mod.def("precond_sqsum_dev", &precond_sqsum_dev,
            "precond_sqsum with device-resident scalars (graph-safe)");
    mod.def("fused_sgd", &fused_sgd, "fused flat-bucket SGD step");
    mod.def("fused_adamw", &fused_adamw, "fused flat-bucket Adam(W) step");
    mod.def("bn_fwd", &bn_fwd, "fused NHWC bf16 BatchNorm(+ReLU) forward");
    mod.def("bn_bwd", &bn_bwd, "fused NHWC bf16 BatchNorm(+ReLU) backward");
    mod.def("conv_wrw_ok", &conv_wrw_ok, "3x3 wrw fast-path predicate");
    mod.def("conv_wrw_nsplit", &conv_wrw_nsplit, "wrw workspace splits");
    mod.def("conv_wrw", &conv_wrw, "MFMA 3x3 s1 NHWC weight gradient");
    mod.def("conv_mm_ok", &conv_mm_ok, "3x3 fwd/bwd-data predicate");
    mod.def("conv_mm", &conv_mm, "MFMA 3x3 s1 NHWC conv (fwd/bwd-data)");
    mod.def("conv_s2_fwd_ok", &conv_s2_fwd_ok,
            "3x3 stride-2 forward predicate (experimental)");
    mod.def("conv_s2_fwd", &conv_s2_fwd,
            "MFMA 3x3 s2 NHWC forward (experimental)");
    mod.def("conv_s2_wrw_ok", &conv_s2_wrw_ok,
            "3x3 stride-2 wrw predicate (experimental)");
    mod.def("conv_s2_wrw_nsplit", &conv_s2_wrw_nsplit,
            "s2 wrw workspace splits");
    mod.def("conv_s2_wrw", &conv_s2_wrw,
            "MFMA 3x3 s2 NHWC weight gradient (experimental)");
    mod.def("conv_s2_bwd_w8b_ok", &conv_s2_bwd_w8b_ok,
            "Wo=8 s2 bwd-data P2=4 redesign predicate");
    mod.def("conv_s2_bwd_w8b", &conv_s2_bwd_w8b,
            "Wo=8 s2 bwd-data P2=4 redesign (experimental)");
    mod.def("conv_s2_bwd_ok", &conv_s2_bwd_ok,
            "3x3 stride-2 bwd-data predicate");
    mod.def("conv_s2_bwd", &conv_s2_bwd,
            "MFMA 3x3 s2 NHWC backward-data (polyphase)");
}
