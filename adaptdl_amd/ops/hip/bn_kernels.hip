// CDNA4 (gfx950 / MI355X) fused NHWC BatchNorm(+ReLU) training kernels.
//
// Replaces MIOpen's spatial batchnorm for channels_last bf16 conv nets:
// rocprof on ResNet-18/CIFAR (profiles/bench_r02_channels_last_kernel_
// stats.csv) shows MIOpen's NHWC BN at 56% of total kernel time, ~10x off
// the HBM3E bandwidth bound.  These kernels are plain bandwidth-bound
// passes designed for the memory-bound rules of
// /opt/skills/guides/cdna_hip_programming.md (Guidelines 11-13, App. B):
//   - bf16 loads/stores vectorized 8-wide (16 B/lane),
//   - per-thread fp32 partials -> LDS tree over the block's row group ->
//     one atomicAdd per (block, channel-octet) (Guideline 12),
//   - grid-stride over rows, grid capped at 2048 blocks,
//   - ReLU (and its backward mask, recomputed from x — no extra read)
//     fused into the BN passes so no separate threshold kernels run.
//
// Layout contract: x is NHWC ("channels_last") bf16 with C % 8 == 0 and
// C/8 <= 256; per-channel parameters are fp32.  M = N*H*W rows.
//
// fwd:  reduce(x) -> sums[2C];  finalize -> mean/rstd/scale/shift +
//       running-stat update;  apply: y = relu?(x*scale+shift)
// bwd:  reduce(x, dy) -> sums[2C] (masked dy, dy*xhat);  finalize ->
//       dgamma/dbeta + per-channel (p, q, r);  apply: dx = p*g - q*x + r

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define BLOCK 256
#define MAX_BLOCKS 2048
#define VEC 8

typedef unsigned short ushort_t;

union Vec8 {
    float4 f4;
    ushort_t u[VEC];
};

__device__ __forceinline__ float b2f(ushort_t h) {
    union { float f; unsigned u; } c;
    c.u = ((unsigned)h) << 16;
    return c.f;
}

__device__ __forceinline__ ushort_t f2b(float f) {
    __hip_bfloat16 h = __float2bfloat16(f);  // RNE
    return *reinterpret_cast<ushort_t*>(&h);
}

// ---------------------------------------------------------------------
// Forward reduction: sums[c] += sum_x, sums[C + c] += sum_x2.
// Threads: slot = tid % slots (owns channels slot*8..+8), row group
// rg = tid / slots; rows grid-stride with stride gridDim.x * rpb.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_fwd_reduce(
        const ushort_t* __restrict__ x, long m, int c,
        float* __restrict__ sums) {
    const int slots = c / VEC;
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    const int slot = threadIdx.x % slots;
    const int rg = threadIdx.x / slots;
    __shared__ float lds[BLOCK * VEC];

    float s[VEC], s2[VEC];
    #pragma unroll
    for (int k = 0; k < VEC; ++k) { s[k] = 0.f; s2[k] = 0.f; }

    if (rg < rpb) {
        const ushort_t* base = x + (size_t)slot * VEC;
        for (long row = (long)blockIdx.x * rpb + rg; row < m;
                row += (long)gridDim.x * rpb) {
            Vec8 v;
            v.f4 = *reinterpret_cast<const float4*>(base + (size_t)row * c);
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float f = b2f(v.u[k]);
                s[k] += f;
                s2[k] += f * f;
            }
        }
    }
    // LDS tree over the row-group dimension (rg), separately for s and s2.
    // Index layout: lds[tid * VEC + k], tid = rg * slots + slot.
    int p2 = 1;
    while (p2 * 2 <= rpb) p2 *= 2;
    #pragma unroll
    for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = s[k];
    __syncthreads();
    if (rg < p2 && rg + p2 < rpb) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            s[k] += lds[(threadIdx.x + p2 * slots) * VEC + k];
    }
    for (int r = p2 / 2; r > 0; r >>= 1) {
        if (rg < p2) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = s[k];
        }
        __syncthreads();
        if (rg < r) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k)
                s[k] += lds[(threadIdx.x + r * slots) * VEC + k];
        }
        __syncthreads();
    }
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            atomicAdd(&sums[slot * VEC + k], s[k]);
    }
    // Second pass of the same tree for s2 (reuse LDS).
    __syncthreads();
    #pragma unroll
    for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = s2[k];
    __syncthreads();
    if (rg < p2 && rg + p2 < rpb) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            s2[k] += lds[(threadIdx.x + p2 * slots) * VEC + k];
    }
    for (int r = p2 / 2; r > 0; r >>= 1) {
        if (rg < p2) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = s2[k];
        }
        __syncthreads();
        if (rg < r) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k)
                s2[k] += lds[(threadIdx.x + r * slots) * VEC + k];
        }
        __syncthreads();
    }
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            atomicAdd(&sums[c + slot * VEC + k], s2[k]);
    }
}

// ---------------------------------------------------------------------
// Forward finalize (C threads): statistics + affine fold + running stats.
// ---------------------------------------------------------------------
extern "C" __global__ void k_bn_fwd_finalize(
        const float* __restrict__ sums, long m, int c,
        const float* __restrict__ gamma, const float* __restrict__ beta,
        float* __restrict__ running_mean, float* __restrict__ running_var,
        float momentum, float eps,
        float* __restrict__ save_mean, float* __restrict__ save_rstd,
        float* __restrict__ scale, float* __restrict__ shift) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= c) return;
    float inv_m = 1.0f / (float)m;
    float mean = sums[i] * inv_m;
    float var = sums[c + i] * inv_m - mean * mean;
    var = var > 0.f ? var : 0.f;
    float rstd = rsqrtf(var + eps);
    save_mean[i] = mean;
    save_rstd[i] = rstd;
    float sc = gamma[i] * rstd;
    scale[i] = sc;
    shift[i] = beta[i] - mean * sc;
    if (running_mean != nullptr) {
        float unbiased = m > 1 ? var * (float)m / (float)(m - 1) : var;
        running_mean[i] += momentum * (mean - running_mean[i]);
        running_var[i] += momentum * (unbiased - running_var[i]);
    }
}

// Eval-mode finalize: fold running statistics into scale/shift.
extern "C" __global__ void k_bn_eval_finalize(
        int c, const float* __restrict__ gamma,
        const float* __restrict__ beta,
        const float* __restrict__ running_mean,
        const float* __restrict__ running_var, float eps,
        float* __restrict__ save_mean, float* __restrict__ save_rstd,
        float* __restrict__ scale, float* __restrict__ shift) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= c) return;
    float mean = running_mean[i];
    float rstd = rsqrtf(running_var[i] + eps);
    save_mean[i] = mean;
    save_rstd[i] = rstd;
    float sc = gamma[i] * rstd;
    scale[i] = sc;
    shift[i] = beta[i] - mean * sc;
}

// ---------------------------------------------------------------------
// Forward apply: y = [relu](x * scale + shift), 8 channels per lane.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_fwd_apply(
        const ushort_t* __restrict__ x, ushort_t* __restrict__ y,
        long nvec, int slots, const float* __restrict__ scale,
        const float* __restrict__ shift, int relu) {
    long i = (long)blockIdx.x * BLOCK + threadIdx.x;
    const long stride = (long)gridDim.x * BLOCK;
    for (long v = i; v < nvec; v += stride) {
        const int cb = (int)(v % slots) * VEC;
        Vec8 in, out;
        in.f4 = *reinterpret_cast<const float4*>(x + v * VEC);
        #pragma unroll
        for (int k = 0; k < VEC; ++k) {
            float f = fmaf(b2f(in.u[k]), scale[cb + k], shift[cb + k]);
            if (relu) f = f > 0.f ? f : 0.f;
            out.u[k] = f2b(f);
        }
        *reinterpret_cast<float4*>(y + v * VEC) = out.f4;
    }
}

// ---------------------------------------------------------------------
// Backward reduction: masked dy and dy * xhat sums per channel.
// g = dy * (relu ? (x*scale+shift > 0) : 1); xhat = (x - mean) * rstd.
// sums[c] += sum g; sums[C + c] += sum g * xhat.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_bwd_reduce(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
        long m, int c, const float* __restrict__ mean,
        const float* __restrict__ rstd, const float* __restrict__ scale,
        const float* __restrict__ shift, int relu,
        float* __restrict__ sums) {
    const int slots = c / VEC;
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    const int slot = threadIdx.x % slots;
    const int rg = threadIdx.x / slots;
    __shared__ float lds[BLOCK * VEC];

    float sg[VEC], sgx[VEC];
    float mn[VEC], rs[VEC], sc[VEC], sh[VEC];
    #pragma unroll
    for (int k = 0; k < VEC; ++k) {
        sg[k] = 0.f; sgx[k] = 0.f;
        mn[k] = mean[slot * VEC + k];
        rs[k] = rstd[slot * VEC + k];
        sc[k] = scale[slot * VEC + k];
        sh[k] = shift[slot * VEC + k];
    }
    if (rg < rpb) {
        const ushort_t* xb = x + (size_t)slot * VEC;
        const ushort_t* db = dy + (size_t)slot * VEC;
        for (long row = (long)blockIdx.x * rpb + rg; row < m;
                row += (long)gridDim.x * rpb) {
            Vec8 vx, vd;
            vx.f4 = *reinterpret_cast<const float4*>(xb + (size_t)row * c);
            vd.f4 = *reinterpret_cast<const float4*>(db + (size_t)row * c);
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float fx = b2f(vx.u[k]);
                float g = b2f(vd.u[k]);
                if (relu && fmaf(fx, sc[k], sh[k]) <= 0.f) g = 0.f;
                sg[k] += g;
                sgx[k] += g * (fx - mn[k]) * rs[k];
            }
        }
    }
    int p2 = 1;
    while (p2 * 2 <= rpb) p2 *= 2;
    #pragma unroll
    for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = sg[k];
    __syncthreads();
    if (rg < p2 && rg + p2 < rpb) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            sg[k] += lds[(threadIdx.x + p2 * slots) * VEC + k];
    }
    for (int r = p2 / 2; r > 0; r >>= 1) {
        if (rg < p2) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = sg[k];
        }
        __syncthreads();
        if (rg < r) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k)
                sg[k] += lds[(threadIdx.x + r * slots) * VEC + k];
        }
        __syncthreads();
    }
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            atomicAdd(&sums[slot * VEC + k], sg[k]);
    }
    __syncthreads();
    #pragma unroll
    for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = sgx[k];
    __syncthreads();
    if (rg < p2 && rg + p2 < rpb) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            sgx[k] += lds[(threadIdx.x + p2 * slots) * VEC + k];
    }
    for (int r = p2 / 2; r > 0; r >>= 1) {
        if (rg < p2) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k) lds[threadIdx.x * VEC + k] = sgx[k];
        }
        __syncthreads();
        if (rg < r) {
            #pragma unroll
            for (int k = 0; k < VEC; ++k)
                sgx[k] += lds[(threadIdx.x + r * slots) * VEC + k];
        }
        __syncthreads();
    }
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            atomicAdd(&sums[c + slot * VEC + k], sgx[k]);
    }
}

// ---------------------------------------------------------------------
// Backward finalize (C threads): dgamma/dbeta + dx coefficients.
// dx = p * g - q * x + r  with g = masked dy:
//   p = gamma * rstd
//   q = p * (sum_gx / M) * rstd          (0 in eval mode)
//   r = -p * (sum_g / M) + q * mean      (0 in eval mode)
// ---------------------------------------------------------------------
extern "C" __global__ void k_bn_bwd_finalize(
        const float* __restrict__ sums, long m, int c,
        const float* __restrict__ gamma, const float* __restrict__ mean,
        const float* __restrict__ rstd, int train,
        float* __restrict__ dgamma, float* __restrict__ dbeta,
        float* __restrict__ pqr) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= c) return;
    float sum_g = sums[i];
    float sum_gx = sums[c + i];
    dbeta[i] = sum_g;
    dgamma[i] = sum_gx;
    float p = gamma[i] * rstd[i];
    float q = 0.f, r = 0.f;
    if (train) {
        float inv_m = 1.0f / (float)m;
        q = p * (sum_gx * inv_m) * rstd[i];
        r = -p * (sum_g * inv_m) + q * mean[i];
    }
    pqr[i] = p;
    pqr[c + i] = q;
    pqr[2 * c + i] = r;
}

// ---------------------------------------------------------------------
// Backward apply: dx = p * g - q * x + r (g = masked dy).
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_bwd_apply(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
        ushort_t* __restrict__ dx, long nvec, int slots, int c,
        const float* __restrict__ scale, const float* __restrict__ shift,
        const float* __restrict__ pqr, int relu) {
    long i = (long)blockIdx.x * BLOCK + threadIdx.x;
    const long stride = (long)gridDim.x * BLOCK;
    for (long v = i; v < nvec; v += stride) {
        const int cb = (int)(v % slots) * VEC;
        Vec8 vx, vd, out;
        vx.f4 = *reinterpret_cast<const float4*>(x + v * VEC);
        vd.f4 = *reinterpret_cast<const float4*>(dy + v * VEC);
        #pragma unroll
        for (int k = 0; k < VEC; ++k) {
            float fx = b2f(vx.u[k]);
            float g = b2f(vd.u[k]);
            if (relu && fmaf(fx, scale[cb + k], shift[cb + k]) <= 0.f)
                g = 0.f;
            float p = pqr[cb + k];
            float q = pqr[c + cb + k];
            float r = pqr[2 * c + cb + k];
            out.u[k] = f2b(fmaf(p, g, fmaf(-q, fx, r)));
        }
        *reinterpret_cast<float4*>(dx + v * VEC) = out.f4;
    }
}

// ---- host-side launchers ---------------------------------------------

static inline unsigned bn_grid(long work_items, int per_block) {
    long blocks = (work_items + per_block - 1) / per_block;
    if (blocks > MAX_BLOCKS) blocks = MAX_BLOCKS;
    if (blocks < 1) blocks = 1;
    return (unsigned)blocks;
}

extern "C" void launch_bn_fwd(
        const ushort_t* x, ushort_t* y, long m, int c,
        const float* gamma, const float* beta, float* running_mean,
        float* running_var, float momentum, float eps, int train, int relu,
        float* sums, float* save_mean, float* save_rstd, float* scale,
        float* shift, hipStream_t s) {
    const int slots = c / VEC;
    if (train) {
        hipMemsetAsync(sums, 0, 2 * c * sizeof(float), s);
        const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
        hipLaunchKernelGGL(k_bn_fwd_reduce, dim3(bn_grid(m, rpb)),
                           dim3(BLOCK), 0, s, x, m, c, sums);
        hipLaunchKernelGGL(k_bn_fwd_finalize,
                           dim3((c + BLOCK - 1) / BLOCK), dim3(BLOCK), 0, s,
                           sums, m, c, gamma, beta, running_mean,
                           running_var, momentum, eps, save_mean, save_rstd,
                           scale, shift);
    } else {
        hipLaunchKernelGGL(k_bn_eval_finalize,
                           dim3((c + BLOCK - 1) / BLOCK), dim3(BLOCK), 0, s,
                           c, gamma, beta, running_mean, running_var, eps,
                           save_mean, save_rstd, scale, shift);
    }
    const long nvec = m * slots;
    hipLaunchKernelGGL(k_bn_fwd_apply, dim3(bn_grid(nvec, BLOCK)),
                       dim3(BLOCK), 0, s, x, y, nvec, slots, scale, shift,
                       relu);
}

extern "C" void launch_bn_bwd(
        const ushort_t* x, const ushort_t* dy, ushort_t* dx, long m, int c,
        const float* gamma, const float* save_mean, const float* save_rstd,
        const float* scale, const float* shift, int train, int relu,
        float* sums, float* dgamma, float* dbeta, float* pqr,
        hipStream_t s) {
    const int slots = c / VEC;
    hipMemsetAsync(sums, 0, 2 * c * sizeof(float), s);
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    hipLaunchKernelGGL(k_bn_bwd_reduce, dim3(bn_grid(m, rpb)), dim3(BLOCK),
                       0, s, x, dy, m, c, save_mean, save_rstd, scale,
                       shift, relu, sums);
    hipLaunchKernelGGL(k_bn_bwd_finalize, dim3((c + BLOCK - 1) / BLOCK),
                       dim3(BLOCK), 0, s, sums, m, c, gamma, save_mean,
                       save_rstd, train, dgamma, dbeta, pqr);
    const long nvec = m * slots;
    hipLaunchKernelGGL(k_bn_bwd_apply, dim3(bn_grid(nvec, BLOCK)),
                       dim3(BLOCK), 0, s, x, dy, dx, nvec, slots, c, scale,
                       shift, pqr, relu);
}
