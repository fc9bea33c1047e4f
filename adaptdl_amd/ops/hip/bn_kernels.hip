// CDNA4 (gfx950 / MI355X) fused NHWC BatchNorm(+ReLU) training kernels.
//
// Replaces MIOpen's spatial batchnorm for channels_last bf16 conv nets:
// rocprof on ResNet-18/CIFAR (profiles/bench_r02_channels_last_kernel_
// stats.csv) shows MIOpen's NHWC BN at 56% of total kernel time, ~10x off
// the HBM3E bandwidth bound.  These kernels are plain bandwidth-bound
// passes designed per /opt/skills/guides/cdna_hip_programming.md
// (Guidelines 11-13, App. B):
//   - bf16 loads/stores vectorized 8-wide (16 B/lane); a wave's 64 lanes
//     cover whole consecutive NHWC rows, so global access is one
//     contiguous 1 KiB span per wave instruction,
//   - every thread owns a FIXED channel octet (slot); per-channel
//     parameters are loaded into registers once, outside the row loop,
//   - reductions are two-stage: per-block partials to a workspace (NO
//     cross-block atomics — v1 used one atomicAdd per (block, channel)
//     and measured 432 us vs the 20 us bandwidth-bound apply pass, pure
//     atomic serialization), then a parallel per-channel block reduce,
//   - ReLU and its backward mask (recomputed from x — no extra stream)
//     are folded in, so no separate threshold/clamp kernels run.
//
// Layout contract: x is NHWC ("channels_last") bf16 with C % 8 == 0 and
// C/8 <= 256; per-channel parameters fp32.  M = N*H*W rows.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define BLOCK 256
#define MAX_RBLOCKS 1024   // stage-1 reduction grid cap (partials rows)
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

// Shared LDS tree over the row-group dimension for VEC per-thread values.
// tid = rg * slots + slot; reduces rg in [0, rpb) down to rg == 0.
__device__ __forceinline__ void rg_tree_reduce(
        float (&s)[VEC], float* lds, int slots, int rpb, int rg) {
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
}

// ---------------------------------------------------------------------
// Stage-1 forward reduction: per-block partial (sum, sumsq) per channel.
// ws layout: ws[c * nb + block] (c in [0, C) sums, [C, 2C) sum-squares)
// so the stage-2 per-channel reduce reads contiguously.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_fwd_reduce(
        const ushort_t* __restrict__ x, long m, int c, int nb,
        float* __restrict__ ws) {
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
                s2[k] = fmaf(f, f, s2[k]);
            }
        }
    }
    rg_tree_reduce(s, lds, slots, rpb, rg);
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            ws[(size_t)(slot * VEC + k) * nb + blockIdx.x] = s[k];
    }
    __syncthreads();
    rg_tree_reduce(s2, lds, slots, rpb, rg);
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            ws[(size_t)(c + slot * VEC + k) * nb + blockIdx.x] = s2[k];
    }
}

// ---------------------------------------------------------------------
// Stage-2: one block per channel-statistic row; sums ws[row * nb .. +nb].
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_reduce_ws(
        const float* __restrict__ ws, int nb, float* __restrict__ out) {
    __shared__ float lds[BLOCK];
    const float* row = ws + (size_t)blockIdx.x * nb;
    float s = 0.f;
    for (int i = threadIdx.x; i < nb; i += BLOCK) s += row[i];
    // wave shuffle reduce, then LDS across the block's 4 waves
    #pragma unroll
    for (int off = WAVE / 2; off > 0; off >>= 1)
        s += __shfl_down(s, off, WAVE);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    if (lane == 0) lds[wid] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.f;
        #pragma unroll
        for (int i = 0; i < BLOCK / WAVE; ++i) t += lds[i];
        out[blockIdx.x] = t;
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
// Forward apply: y = [relu](x * scale + shift).  Row-loop structure with
// a fixed channel octet per thread so scale/shift live in registers.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_fwd_apply(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ z,
        ushort_t* __restrict__ y, long m, int c,
        const float* __restrict__ scale,
        const float* __restrict__ shift, int relu) {
    const int slots = c / VEC;
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    const int slot = threadIdx.x % slots;
    const int rg = threadIdx.x / slots;
    if (rg >= rpb) return;
    float sc[VEC], sh[VEC];
    #pragma unroll
    for (int k = 0; k < VEC; ++k) {
        sc[k] = scale[slot * VEC + k];
        sh[k] = shift[slot * VEC + k];
    }
    const ushort_t* xb = x + (size_t)slot * VEC;
    const ushort_t* zb = z ? z + (size_t)slot * VEC : nullptr;
    ushort_t* yb = y + (size_t)slot * VEC;
    for (long row = (long)blockIdx.x * rpb + rg; row < m;
            row += (long)gridDim.x * rpb) {
        Vec8 in, vz, out;
        in.f4 = *reinterpret_cast<const float4*>(xb + (size_t)row * c);
        if (zb)
            vz.f4 = *reinterpret_cast<const float4*>(zb + (size_t)row * c);
        #pragma unroll
        for (int k = 0; k < VEC; ++k) {
            float f = fmaf(b2f(in.u[k]), sc[k], sh[k]);
            if (zb) f += b2f(vz.u[k]);
            if (relu) f = f > 0.f ? f : 0.f;
            out.u[k] = f2b(f);
        }
        *reinterpret_cast<float4*>(yb + (size_t)row * c) = out.f4;
    }
}

// ---------------------------------------------------------------------
// Stage-1 backward reduction: per-block partials of (masked dy) and
// (masked dy * xhat) per channel, into the same ws layout as forward.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_bwd_reduce(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ z,
        const ushort_t* __restrict__ dy,
        long m, int c, int nb, const float* __restrict__ mean,
        const float* __restrict__ rstd, const float* __restrict__ scale,
        const float* __restrict__ shift, int relu,
        float* __restrict__ ws) {
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
        const ushort_t* zb = z ? z + (size_t)slot * VEC : nullptr;
        const ushort_t* db = dy + (size_t)slot * VEC;
        for (long row = (long)blockIdx.x * rpb + rg; row < m;
                row += (long)gridDim.x * rpb) {
            Vec8 vx, vz, vd;
            vx.f4 = *reinterpret_cast<const float4*>(xb + (size_t)row * c);
            if (zb)
                vz.f4 = *reinterpret_cast<const float4*>(
                    zb + (size_t)row * c);
            vd.f4 = *reinterpret_cast<const float4*>(db + (size_t)row * c);
            #pragma unroll
            for (int k = 0; k < VEC; ++k) {
                float fx = b2f(vx.u[k]);
                float g = b2f(vd.u[k]);
                if (relu) {
                    float pre = fmaf(fx, sc[k], sh[k]);
                    if (zb) pre += b2f(vz.u[k]);
                    if (pre <= 0.f) g = 0.f;
                }
                sg[k] += g;
                sgx[k] = fmaf(g, (fx - mn[k]) * rs[k], sgx[k]);
            }
        }
    }
    rg_tree_reduce(sg, lds, slots, rpb, rg);
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            ws[(size_t)(slot * VEC + k) * nb + blockIdx.x] = sg[k];
    }
    __syncthreads();
    rg_tree_reduce(sgx, lds, slots, rpb, rg);
    if (rg == 0) {
        #pragma unroll
        for (int k = 0; k < VEC; ++k)
            ws[(size_t)(c + slot * VEC + k) * nb + blockIdx.x] = sgx[k];
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
// Backward apply: dx = p * g - q * x + r (g = masked dy), row-loop with
// register-resident per-channel coefficients.
// ---------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(BLOCK) void k_bn_bwd_apply(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ z,
        const ushort_t* __restrict__ dy,
        ushort_t* __restrict__ dx, ushort_t* __restrict__ dz, long m,
        int c, const float* __restrict__ scale,
        const float* __restrict__ shift,
        const float* __restrict__ pqr, int relu) {
    const int slots = c / VEC;
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    const int slot = threadIdx.x % slots;
    const int rg = threadIdx.x / slots;
    if (rg >= rpb) return;
    float sc[VEC], sh[VEC], cp[VEC], cq[VEC], cr[VEC];
    #pragma unroll
    for (int k = 0; k < VEC; ++k) {
        sc[k] = scale[slot * VEC + k];
        sh[k] = shift[slot * VEC + k];
        cp[k] = pqr[slot * VEC + k];
        cq[k] = pqr[c + slot * VEC + k];
        cr[k] = pqr[2 * c + slot * VEC + k];
    }
    const ushort_t* xb = x + (size_t)slot * VEC;
    const ushort_t* zb = z ? z + (size_t)slot * VEC : nullptr;
    const ushort_t* db = dy + (size_t)slot * VEC;
    ushort_t* ob = dx + (size_t)slot * VEC;
    ushort_t* zo = dz ? dz + (size_t)slot * VEC : nullptr;
    for (long row = (long)blockIdx.x * rpb + rg; row < m;
            row += (long)gridDim.x * rpb) {
        Vec8 vx, vz, vd, out, outz;
        vx.f4 = *reinterpret_cast<const float4*>(xb + (size_t)row * c);
        if (zb)
            vz.f4 = *reinterpret_cast<const float4*>(zb + (size_t)row * c);
        vd.f4 = *reinterpret_cast<const float4*>(db + (size_t)row * c);
        #pragma unroll
        for (int k = 0; k < VEC; ++k) {
            float fx = b2f(vx.u[k]);
            float g = b2f(vd.u[k]);
            if (relu) {
                float pre = fmaf(fx, sc[k], sh[k]);
                if (zb) pre += b2f(vz.u[k]);
                if (pre <= 0.f) g = 0.f;
            }
            if (zo) outz.u[k] = f2b(g);
            out.u[k] = f2b(fmaf(cp[k], g, fmaf(-cq[k], fx, cr[k])));
        }
        *reinterpret_cast<float4*>(ob + (size_t)row * c) = out.f4;
        if (zo)
            *reinterpret_cast<float4*>(zo + (size_t)row * c) = outz.f4;
    }
}

// ---- host-side launchers ---------------------------------------------

static inline unsigned bn_grid(long work_items, int per_block, long cap) {
    long blocks = (work_items + per_block - 1) / per_block;
    if (blocks > cap) blocks = cap;
    if (blocks < 1) blocks = 1;
    return (unsigned)blocks;
}

extern "C" void launch_bn_fwd(
        const ushort_t* x, const ushort_t* z, ushort_t* y, long m, int c,
        const float* gamma, const float* beta, float* running_mean,
        float* running_var, float momentum, float eps, int train, int relu,
        float* ws, float* sums, float* save_mean, float* save_rstd,
        float* scale, float* shift, hipStream_t s) {
    const int slots = c / VEC;
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    if (train) {
        const unsigned nb = bn_grid(m, rpb, MAX_RBLOCKS);
        hipLaunchKernelGGL(k_bn_fwd_reduce, dim3(nb), dim3(BLOCK), 0, s,
                           x, m, c, (int)nb, ws);
        hipLaunchKernelGGL(k_bn_reduce_ws, dim3(2 * c), dim3(BLOCK), 0, s,
                           ws, (int)nb, sums);
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
    hipLaunchKernelGGL(k_bn_fwd_apply, dim3(bn_grid(m, rpb, MAX_BLOCKS)),
                       dim3(BLOCK), 0, s, x, z, y, m, c, scale, shift,
                       relu);
}

extern "C" void launch_bn_bwd(
        const ushort_t* x, const ushort_t* z, const ushort_t* dy,
        ushort_t* dx, ushort_t* dz, long m, int c,
        const float* gamma, const float* save_mean, const float* save_rstd,
        const float* scale, const float* shift, int train, int relu,
        float* ws, float* sums, float* dgamma, float* dbeta, float* pqr,
        hipStream_t s) {
    const int slots = c / VEC;
    const int rpb = BLOCK / slots > 0 ? BLOCK / slots : 1;
    const unsigned nb = bn_grid(m, rpb, MAX_RBLOCKS);
    hipLaunchKernelGGL(k_bn_bwd_reduce, dim3(nb), dim3(BLOCK), 0, s,
                       x, z, dy, m, c, (int)nb, save_mean, save_rstd,
                       scale, shift, relu, ws);
    hipLaunchKernelGGL(k_bn_reduce_ws, dim3(2 * c), dim3(BLOCK), 0, s,
                       ws, (int)nb, sums);
    hipLaunchKernelGGL(k_bn_bwd_finalize, dim3((c + BLOCK - 1) / BLOCK),
                       dim3(BLOCK), 0, s, sums, m, c, gamma, save_mean,
                       save_rstd, train, dgamma, dbeta, pqr);
    hipLaunchKernelGGL(k_bn_bwd_apply, dim3(bn_grid(m, rpb, MAX_BLOCKS)),
                       dim3(BLOCK), 0, s, x, z, dy, dx, dz, m, c, scale,
                       shift, pqr, relu);
}
