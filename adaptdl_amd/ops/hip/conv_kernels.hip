// CDNA4 (gfx950) MFMA conv kernels: 3x3 stride-1 NHWC weight gradient.
//
// MIOpen's igemm wrw kernels are the single worst hot spot of the
// flagship ResNet workload (profiles/bench_r03: 256 us where the MFMA
// floor is ~33 us, plus SubTensorOp zero/cast fills).  This kernel
// computes dW[k][tau][c] = sum_{n,h,w} dy[n,h,w,k] * x[n,h+dh,w+dw,c]
// as an implicit GEMM with v_mfma_f32_16x16x32_bf16:
//
//   - a "chunk" is P consecutive h-lines of one image (P*W pixels, a
//     multiple of 32 = the MFMA K dimension),
//   - dy and x chunk tiles are staged TRANSPOSED in LDS
//     (dy_t[k][pixel], x_t[c][line-padded pixel]) so both MFMA operands
//     read contiguous 16-byte fragments (ds_read_b128),
//   - x lines carry left/right zero pads and halo lines, so all nine
//     (dh, dw) taps come from two ALIGNED b128 reads per (dh, c-frag)
//     window plus compile-time byte-rotations for dw = 0, +1
//     (w-offsets are multiples of 8 by construction),
//   - each wave owns a 32k x 32c output tile: 2x2 fragments x 9 taps =
//     36 fp32 accumulators fragments; a 4-wave block owns 64k x 64c,
//   - grid = (K/64) * (C/64) * SPLIT blocks; each split accumulates its
//     share of chunks and writes one fp32 partial slab; a second kernel
//     reduces the slabs (no cross-block atomics - see bn_kernels.hip
//     for the measured atomic-serialization lesson).
//
// MFMA operand maps verified on hardware by tools/mfma_probe.hip:
//   A[m][k]: lane l elem e -> m = l%16, k = (l/16)*8+e
//   B[k][n]: lane l elem e -> k = (l/16)*8+e, n = l%16
//   D[m][n]: lane l reg  r -> m = (l>>4)*4+r, n = l&15
//
// Constraints: 3x3, stride 1, pad 1, dilation 1, groups 1, NHWC bf16,
// C % 64 == 0, K % 64 == 0, W in {8, 16, 32}, H % P == 0.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdlib>

#define WAVE 64

__device__ __forceinline__ unsigned short f2b(float f) {
    __hip_bfloat16 h = __float2bfloat16(f);  // RNE
    return *reinterpret_cast<unsigned short*>(&h);
}

typedef unsigned short ushort_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

// Strides + a 32-byte-granule XOR swizzle chosen by offline search
// against gfx950's ACTUAL ds_read_b128 lane groups (four non-contiguous
// 16-lane sets, MI355X_MICROARCH §LDS) and ds_write b32 halves:
// modeled conflict cycles drop 3x vs the naive 136/248 strides.
#define DY_STRIDE 160        // shorts per dy_t row
#define XT_STRIDE 288        // shorts per x_t row
// Column-only 32-byte-granule XOR (a per-row bijection): offline search
// against the real b128 lane groups drops modeled conflict cycles from
// 512/640 to 64/64 for the dy/x tiles at these strides.
#define DSWZ(row, col) (((row) * 0) + ((col) ^ ((((row) >> 3) & 7) << 4)))
#define CPMAX 128

union V16 {
    u32x4 u4;
    unsigned int u[4];
    short s[8];
};

// Software-pipelined (T14): two LDS buffer sets; the global loads for
// chunk i+1 are issued BEFORE chunk i's MFMA phase and land in LDS
// after it, so HBM latency hides under compute.  v3 measured wall time
// ~5x the per-SIMD busy cycles at 1 block/CU -- pure staging latency.

struct WrwRegs {
    V16 vdy[1][2];
    V16 vx[2][2];
};

// Wp is the padded line width (next multiple of 8 >= W): pixels with
// w in [W, Wp) and tail-chunk lines with h >= H stage ZERO dy, so they
// contribute nothing to the accumulation — this is what lets the
// ImageNet-resolution widths (56/28/14/7) run on the same MFMA tiling
// (legacy W in {8,16,32} has Wp == W and the guards are always true).
__device__ __forceinline__ void wrw_issue(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
        int n, int h0, int H, int W, int Wp, int C, int K, int kt, int ct,
        int CP, int P, int t, WrwRegs& r) {
#ifdef WRW_PROBE_NOLOAD
    (void)x; (void)dy; (void)n; (void)h0;
    return;
#endif
    const int dyn8 = CP * 4;
    #pragma unroll
    for (int it = 0; it < 1; ++it) {
        const int i = t + it * 512;
        if (i < dyn8) {
            const int p = (i >> 3) * 2;
            const int h = h0 + p / Wp;
            const int w = p % Wp;
            const ushort_t* g = dy +
                (((size_t)n * H + h) * W + w) * K
                + (size_t)kt * 64 + (i & 7) * 8;
            if (h < H && w < W)
                r.vdy[it][0].u4 = *reinterpret_cast<const u32x4*>(g);
            else
                r.vdy[it][0].u4 = u32x4{0u, 0u, 0u, 0u};
            if (h < H && w + 1 < W)
                r.vdy[it][1].u4 = *reinterpret_cast<const u32x4*>(g + K);
            else
                r.vdy[it][1].u4 = u32x4{0u, 0u, 0u, 0u};
        }
    }
    const int xn8 = (P + 2) * Wp * 4;
    #pragma unroll
    for (int it = 0; it < 2; ++it) {
        const int i = t + it * 512;
        if (i < xn8) {
            const int j = i / (Wp * 4);
            const int rem = i % (Wp * 4);
            const int w = (rem >> 3) * 2;
            const int h = h0 - 1 + j;
            const ushort_t* g = x +
                (((size_t)n * H + h) * W + w) * C
                + (size_t)ct * 64 + (rem & 7) * 8;
            const bool hok = h >= 0 && h < H;
            if (hok && w < W)
                r.vx[it][0].u4 = *reinterpret_cast<const u32x4*>(g);
            else
                r.vx[it][0].u4 = u32x4{0u, 0u, 0u, 0u};
            if (hok && w + 1 < W)
                r.vx[it][1].u4 = *reinterpret_cast<const u32x4*>(g + C);
            else
                r.vx[it][1].u4 = u32x4{0u, 0u, 0u, 0u};
        }
    }
}

__device__ __forceinline__ void wrw_write(
        short* __restrict__ dy_t, short* __restrict__ x_t,
        int Wp, int CP, int P, int LS, int t, WrwRegs& r) {
    const int dyn8 = CP * 4;
    #pragma unroll
    for (int it = 0; it < 1; ++it) {
        const int i = t + it * 512;
        if (i < dyn8) {
            const int p = (i >> 3) * 2;
            const int kg = (i & 7) * 8;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const unsigned int packed =
                    (unsigned int)(unsigned short)r.vdy[it][0].s[j]
                    | ((unsigned int)(unsigned short)
                       r.vdy[it][1].s[j] << 16);
                *reinterpret_cast<unsigned int*>(
                    &dy_t[(kg + j) * DY_STRIDE
                          + DSWZ(kg + j, p)]) = packed;
            }
        }
    }
    const int xn8 = (P + 2) * Wp * 4;
    #pragma unroll
    for (int it = 0; it < 2; ++it) {
        const int i = t + it * 512;
        if (i < xn8) {
            const int j = i / (Wp * 4);
            const int rem = i % (Wp * 4);
            const int w = (rem >> 3) * 2;
            const int cg = (rem & 7) * 8;
            #pragma unroll
            for (int jj = 0; jj < 8; ++jj) {
                const unsigned int packed =
                    (unsigned int)(unsigned short)r.vx[it][0].s[jj]
                    | ((unsigned int)(unsigned short)
                       r.vx[it][1].s[jj] << 16);
                *reinterpret_cast<unsigned int*>(
                    &x_t[(cg + jj) * XT_STRIDE
                         + DSWZ(cg + jj, j * LS + 4 + w)]) = packed;
            }
        }
    }
}

#define WRW_BUF (64 * DY_STRIDE + 64 * XT_STRIDE)

// Compile-time tau subsets per wave pair: WT2=0 -> taus 0..4 (groups
// (dh0, dw0..2), (dh1, dw0..1)); WT2=1 -> taus 5..8 ((dh1, dw2),
// (dh2, dw0..2)).  Everything constant-folds so acc stays in registers.
template <int WT2>
struct WrwTaus {
    static constexpr int NG = 2;
    static constexpr int DH[2] = {WT2 ? 1 : 0, WT2 ? 2 : 1};
    static constexpr int LO[2] = {WT2 ? 2 : 0, 0};
    static constexpr int HI[2] = {3, WT2 ? 3 : 2};
};

template <int WT2>
__device__ __forceinline__ void wrw_mfma_phase(
        const short* __restrict__ dy_t, const short* __restrict__ x_t,
        int kchunks, int wshift, int LS, int wk, int wc, int row16,
        int slot8, f32x4 (&acc)[2][2][5]) {
    for (int kc = 0; kc < kchunks; ++kc) {
        const int p0 = kc * 32 + slot8;
        const int li = p0 >> wshift;           // Wp is a power of two
        const int w0 = p0 & ((1 << wshift) - 1);
        bf16x8 afrag[2];
        #pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            const int row = wk * 32 + mf * 16 + row16;
            const short* a = &dy_t[row * DY_STRIDE
                                   + DSWZ(row, kc * 32 + slot8)];
            afrag[mf] = *reinterpret_cast<const bf16x8*>(a);
        }
        int ti = 0;
        #pragma unroll
        for (int g = 0; g < 2; ++g) {
            constexpr int dh0 = WT2 ? 1 : 0;
            constexpr int dh1 = WT2 ? 2 : 1;
            const int dh = g ? dh1 : dh0;
            constexpr int lo0 = WT2 ? 2 : 0;
            constexpr int hi1 = WT2 ? 3 : 2;
            const int lo = g ? 0 : lo0;
            const int hi = g ? hi1 : 3;
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf) {
                const int crow = wc * 32 + nf * 16 + row16;
                const int cbase = (li + dh) * LS + w0;
                V16 vlo, vhi;
                vlo.u4 = *reinterpret_cast<const u32x4*>(
                    &x_t[crow * XT_STRIDE + DSWZ(crow, cbase)]);
                vhi.u4 = *reinterpret_cast<const u32x4*>(
                    &x_t[crow * XT_STRIDE + DSWZ(crow, cbase + 8)]);
                unsigned int win[8] = {vlo.u[0], vlo.u[1], vlo.u[2],
                                       vlo.u[3], vhi.u[0], vhi.u[1],
                                       vhi.u[2], vhi.u[3]};
                int tj = ti;
                #pragma unroll
                for (int dw = 0; dw < 3; ++dw) {
                    if (dw < lo || dw >= hi) continue;
                    u32x4 frag;
                    const int sh = (3 + dw) * 2;
                    const int d0 = sh >> 2;
                    const int rem = sh & 3;
                    if (rem == 0) {
                        frag[0] = win[d0]; frag[1] = win[d0 + 1];
                        frag[2] = win[d0 + 2]; frag[3] = win[d0 + 3];
                    } else {
                        frag[0] = __builtin_amdgcn_alignbyte(
                            win[d0 + 1], win[d0], rem);
                        frag[1] = __builtin_amdgcn_alignbyte(
                            win[d0 + 2], win[d0 + 1], rem);
                        frag[2] = __builtin_amdgcn_alignbyte(
                            win[d0 + 3], win[d0 + 2], rem);
                        frag[3] = __builtin_amdgcn_alignbyte(
                            win[d0 + 4], win[d0 + 3], rem);
                    }
                    const bf16x8 bfrag =
                        *reinterpret_cast<const bf16x8*>(&frag);
                    #pragma unroll
                    for (int mf = 0; mf < 2; ++mf)
                        acc[mf][nf][tj] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                afrag[mf], bfrag, acc[mf][nf][tj],
                                0, 0, 0);
                    ++tj;
                }
            }
            ti += (g ? hi1 : 3) - (g ? 0 : lo0);
        }
    }
}

// 8 waves: (wk, wc) tile quadrant x 2-wave tau split -- two waves per
// SIMD interleave the alignbyte/window VALU work under each other's
// MFMAs (phase probe: MFMA+VALU phase was ~150 us of the 218 us call
// at 1 wave/SIMD; staging only ~35 us).
extern "C" __global__ __launch_bounds__(512, 2) void k_conv3x3_wrw(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
        float* __restrict__ ws, int N, int H, int W, int Wp, int C, int K,
        int P, int nsplit) {
    __shared__ short lds[2 * WRW_BUF];

    const int LS = Wp + 8;           // padded line stride (mult of 8)
    const int CP = P * Wp;           // chunk pixels (mult of 32)
    const int kchunks = CP / 32;
    const int wshift = 31 - __clz(Wp);
    const int nc = C / 64;
    const int nsplit_t = nsplit;
    const int tile = blockIdx.x / nsplit_t;
    const int split = blockIdx.x % nsplit_t;
    const int kt = tile / nc;
    const int ct = tile % nc;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;
    const int wk = wid >> 2;
    const int wc = (wid >> 1) & 1;
    const int wt2 = wid & 1;         // tau half: 0 -> taus 0..4, 1 -> 5..8
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;

    f32x4 acc[2][2][5];
    #pragma unroll
    for (int mf = 0; mf < 2; ++mf)
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf)
            #pragma unroll
            for (int ti = 0; ti < 5; ++ti)
                #pragma unroll
                for (int r = 0; r < 4; ++r)
                    acc[mf][nf][ti][r] = 0.f;

    const int lines_per_img = (H + P - 1) / P;  // tail chunks zero-fill
    const long chunks_total = (long)N * lines_per_img;

    // Zero the pad columns of BOTH x_t buffers once.
    for (int b = 0; b < 2; ++b) {
        short* x_t = lds + b * WRW_BUF + 64 * DY_STRIDE;
        for (int i = t; i < (P + 2) * 8 * 64; i += 512) {
            const int cc = i / ((P + 2) * 8);
            const int rem = i % ((P + 2) * 8);
            const int j = rem / 8;
            const int pp = rem % 8;
            const int q = pp < 4 ? pp : (4 + Wp + (pp - 4));
            x_t[cc * XT_STRIDE + DSWZ(cc, j * LS + q)] = 0;
        }
    }

    WrwRegs regs;
    #define WRW_NH(qq) \
        const int n_ = (int)((qq) / lines_per_img); \
        const int h0_ = (int)((qq) % lines_per_img) * P;

    // Prologue: stage chunk 0 into buffer 0, issue chunk 1's loads.
    if (split < chunks_total) {
        WRW_NH(split)
        wrw_issue(x, dy, n_, h0_, H, W, Wp, C, K, kt, ct, CP, P, t, regs);
        wrw_write(lds, lds + 64 * DY_STRIDE, Wp, CP, P, LS, t, regs);
    }
    __syncthreads();
    if (split + nsplit_t < chunks_total) {
        WRW_NH(split + nsplit_t)
        wrw_issue(x, dy, n_, h0_, H, W, Wp, C, K, kt, ct, CP, P, t, regs);
    }

    int cur = 0;
    for (long q = split; q < chunks_total; q += nsplit_t, cur ^= 1) {
        const short* dy_t = lds + cur * WRW_BUF;
        const short* x_t = dy_t + 64 * DY_STRIDE;

        // ---- MFMA phase (chunk q, buffer cur) ----
        if (wt2 == 0)
            wrw_mfma_phase<0>(dy_t, x_t, kchunks, wshift, LS, wk, wc,
                              row16, slot8, acc);
        else
            wrw_mfma_phase<1>(dy_t, x_t, kchunks, wshift, LS, wk, wc,
                              row16, slot8, acc);

        // ---- stage chunk q+1 into the other buffer; issue q+2 ----
        if (q + nsplit_t < chunks_total) {
            short* ndy = lds + (cur ^ 1) * WRW_BUF;
            wrw_write(ndy, ndy + 64 * DY_STRIDE, Wp, CP, P, LS, t,
                      regs);
            if (q + 2 * nsplit_t < chunks_total) {
                WRW_NH(q + 2 * nsplit_t)
                wrw_issue(x, dy, n_, h0_, H, W, Wp, C, K, kt, ct, CP, P, t,
                          regs);
            }
        }
        __syncthreads();
    }
    #undef WRW_NH

    // ---- epilogue: write the fp32 partial slab (this wave's taus) ---
    float* slab = ws + (size_t)split * K * 9 * C;
    #pragma unroll
    for (int ti = 0; ti < 5; ++ti) {
        if (wt2 && ti >= 4)
            continue;
        const int tau = (wt2 ? 5 : 0) + ti;
        #pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf) {
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int klocal = wk * 32 + mf * 16
                        + ((lane >> 4) * 4 + r);
                    const int clocal = wc * 32 + nf * 16
                        + (lane & 15);
                    slab[((size_t)(kt * 64 + klocal) * 9 + tau) * C
                         + ct * 64 + clocal] = acc[mf][nf][ti][r];
                }
            }
        }
    }
}

// dW[i] = sum_s ws[s * n + i].  Hierarchical: each block sums a
// <=64-split group over a 1024-float i-chunk (f32x4 per thread), then
// atomically adds into dw (zeroed by the launcher when several groups).
extern "C" __global__ __launch_bounds__(256) void k_wrw_reduce(
        const float* __restrict__ ws, float* __restrict__ dw, long n,
        int nsplit, int ngroups, long nchunks) {
    const long work = nchunks * ngroups;
    for (long b = blockIdx.x; b < work; b += gridDim.x) {
        const long ic = b % nchunks;
        const int sg = (int)(b / nchunks);
        const long i0 = ic * 1024 + (long)threadIdx.x * 4;
        if (i0 >= n) continue;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        const int s_end = min(nsplit, (sg + 1) * 16);
        const bool tail = i0 + 4 > n;
        for (int sp = sg * 16; sp < s_end; ++sp) {
            const float* src = ws + (size_t)sp * n + i0;
            if (!tail) {
                f32x4 v = *reinterpret_cast<const f32x4*>(src);
                acc[0] += v[0]; acc[1] += v[1];
                acc[2] += v[2]; acc[3] += v[3];
            } else {
                for (long j = 0; i0 + j < n; ++j) acc[j] += src[j];
            }
        }
        if (ngroups == 1) {
            if (!tail) {
                *reinterpret_cast<f32x4*>(dw + i0) = acc;
            } else {
                for (long j = 0; i0 + j < n; ++j) dw[i0 + j] = acc[j];
            }
        } else {
            for (long j = 0; j < 4 && i0 + j < n; ++j)
                atomicAdd(&dw[i0 + j], acc[j]);
        }
    }
}

// ---- host-side launchers ---------------------------------------------

// Per-shape chunking: Wp = padded line width (power of two >= W), P =
// lines per chunk; CP = P * Wp <= 128 must be a multiple of 32.  The
// CIFAR widths use full-width chunks (Wp == W); the ImageNet-resolution
// widths (ResNet-50 at 224: 56/28/14/7) zero-pad the line to Wp and
// zero-fill tail-chunk lines, trading <= 24% staged-pixel waste for the
// same aligned-b128 LDS tiling.
extern "C" int conv3x3_wrw_params(int H, int W, int* P, int* Wp) {
    // W=14/7 chunking A/B (r2 pass e, N256 C=K=256/512): the
    // exact-cover small chunks (P=2/P=4, CP=32, 87% pixel utilization)
    // ran 163/183 us - barrier-round-bound at kchunks=1 - while P=8
    // (CP=64/128, 77% utilization via zero-filled tail lines) runs
    // 120/127 us and beats MIOpen's 135/143 us.  Large chunks are the
    // default; ADAPTDL_WRW_SMALLCHUNK restores the exact-cover variant
    // for re-measurement.
    const bool small = getenv("ADAPTDL_WRW_SMALLCHUNK") != nullptr;
    switch (W) {
    case 8:  *Wp = 8;  *P = 8; break;
    case 16: *Wp = 16; *P = 8; break;
    case 32: *Wp = 32; *P = 4; break;
    case 56: *Wp = 64; *P = 1; break;
    case 28: *Wp = 32; *P = 4; break;
    case 14: *Wp = 16; *P = small ? 2 : 8; break;
    case 7:  *Wp = 8;  *P = small ? 4 : 8; break;
    default: return 0;
    }
    if (H < 2) return 0;
    return 1;
}

extern "C" int conv3x3_wrw_supported(int H, int W, int C, int K) {
    if (C % 64 || K % 64) return 0;
    int P, Wp;
    return conv3x3_wrw_params(H, W, &P, &Wp);
}

extern "C" int conv3x3_wrw_nsplit(int N, int H, int W, int C, int K) {
    int P = 8, Wp = 8;
    conv3x3_wrw_params(H, W, &P, &Wp);
    long chunks = (long)N * ((H + P - 1) / P);
    int tiles = (K / 64) * (C / 64);
    long target = 256 / tiles;  // one residency round (1 block/CU at 98 KB LDS)
    if (target < 1) target = 1;
    if (target > chunks) target = chunks;
    return (int)target;
}

extern "C" void launch_conv3x3_wrw(
        const ushort_t* x, const ushort_t* dy, float* ws, float* dw,
        int N, int H, int W, int C, int K, hipStream_t s) {
    int P = 8, Wp = 8;
    conv3x3_wrw_params(H, W, &P, &Wp);
    const int nsplit = conv3x3_wrw_nsplit(N, H, W, C, K);
    const int tiles = (K / 64) * (C / 64);
    hipLaunchKernelGGL(k_conv3x3_wrw, dim3(tiles * nsplit), dim3(512), 0,
                       s, x, dy, ws, N, H, W, Wp, C, K, P, nsplit);
    const long n = (long)K * 9 * C;
    // 16-slab groups: enough blocks to fill the chip (the 64-slab
    // version ran 144 blocks on layer1 and measured 5x off bandwidth).
    const int ngroups = (nsplit + 15) / 16;
    const long nchunks = (n + 1023) / 1024;
    if (ngroups > 1)
        hipMemsetAsync(dw, 0, n * sizeof(float), s);
    long blocks = nchunks * ngroups;
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(k_wrw_reduce, dim3((unsigned)blocks), dim3(256), 0,
                       s, ws, dw, n, nsplit, ngroups, nchunks);
}

// =====================================================================
// MFMA 3x3 stride-1 NHWC convolution: forward AND backward-data.
//
//   y[p][ko] = sum_{tau, ci} x[p + off(tau)][ci] * w[ko][tau][ci]
//
// Backward-data is the same contraction with x := dy, w := the
// flipped/transposed weight ([C][tau'][K], prepared host-side), so one
// kernel serves both directions.  Unlike the wrw kernel, both operands
// stage in their NATURAL layouts (zero transpose cost):
//   - x_s: pixel-major padded window [q][C+pad]; b128 copies, the MFMA
//     A-fragment (16 pixels x 32 c-slots) reads 16B rows directly, with
//     the 9 taps being pure row/column offsets into the padded window,
//   - w_lds: [ko][tau][ci+pad], LDS-resident for the whole block
//     (staged once, reused across the block's chunk loop),
//   - D tile = y pixels x ko, written straight out (no split-K, no
//     workspace: the contraction (9*C <= 1152) fits one block).
// Dynamic LDS (up to ~128 KB) sized host-side.
// Constraints: 3x3 s1 p1, NHWC bf16, W in {16, 32}, H % P == 0,
// C % 32 == 0, C <= 128, K % KT == 0 (KT = 64 for C<=64 else 32).
// =====================================================================

struct MmRegs {
    V16 v[5];
};

#ifdef MM_PROBE_NOSTORE
__device__ bool ybase_guard(float v) { return v != 12345.678f; }
#endif

__device__ __forceinline__ void mm_issue(
        const ushort_t* __restrict__ x, int n, int h0, int H, int W,
        int C, int ct64, int P, int t, MmRegs& r) {
    const int pieces = (P + 2) * W * (C / 8);
    #pragma unroll
    for (int it = 0; it < 5; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (W * (C / 8));
            const int rem = i % (W * (C / 8));
            const int ww = rem / (C / 8);
            const int cg = (rem % (C / 8)) * 8;
            const int h = h0 - 1 + j;
            if (h < 0 || h >= H) {
                r.v[it].u4 = u32x4{0u, 0u, 0u, 0u};
            } else {
                r.v[it].u4 = *reinterpret_cast<const u32x4*>(
                    x + (((size_t)n * H + h) * W + ww) * C + cg);
            }
        }
    }
}

__device__ __forceinline__ void mm_write(
        short* __restrict__ x_s, int W, int C, int CS, int LP, int P,
        int t, MmRegs& r) {
    const int pieces = (P + 2) * W * (C / 8);
    #pragma unroll
    for (int it = 0; it < 5; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (W * (C / 8));
            const int rem = i % (W * (C / 8));
            const int ww = rem / (C / 8);
            const int cg = (rem % (C / 8)) * 8;
            *reinterpret_cast<u32x4*>(
                &x_s[(j * LP + 1 + ww) * CS + cg]) = r.v[it].u4;
        }
    }
}

extern "C" __global__ __launch_bounds__(512, 2) void k_conv3x3_mm(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ w,
        ushort_t* __restrict__ y, int N, int H, int W, int C, int K,
        int P, int KT, int spread, int dbuf) {
    extern __shared__ short lds[];
    const int CS = C + 16;           // x stride: zero read conflicts
    const int WS = C + 8;            // w stride (smaller: LDS budget)
    const int LP = W + 2;            // padded line width (pixels)
    const int QR = (P + 2) * LP;     // x window rows
    short* w_lds = lds;              // [KT * 9][WS]
    short* x_s0 = lds + KT * 9 * WS; // [QR][CS] buffer 0
    short* x_s1 = x_s0 + (dbuf ? QR * CS : 0);  // buffer 1 (pipelined)

    const int CP = P * W;            // chunk pixels (= 128)
    const int nk = K / KT;
    const int kt = blockIdx.x / spread;
    const int sp = blockIdx.x % spread;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;
    const int wp = wid >> 1;         // pixel quarter (0..3): 32 pixels
    const int wk = wid & 1;          // ko half (0..1): KT/2 channels
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;
    const int KH = KT / 2;           // per-wave ko span (32 or 16)
    const int NF = KH / 16;          // ko fragments per wave (2 or 1)

    // ---- stage the weight tile once per block ----
    {
        const int pieces = KT * 9 * (C / 8);
        const ushort_t* wg = w + (size_t)kt * KT * 9 * C;
        for (int i = t; i < pieces; i += 512) {
            const int row = i / (C / 8);        // ko * 9 + tau
            const int cg = (i % (C / 8)) * 8;
            V16 v;
            v.u4 = *reinterpret_cast<const u32x4*>(
                wg + (size_t)row * C + cg);
            *reinterpret_cast<u32x4*>(&w_lds[row * WS + cg]) = v.u4;
        }
    }
    // ---- zero x pad columns of both buffers once ----
    for (int b = 0; b < (dbuf ? 2 : 1); ++b) {
        short* x_s = b ? x_s1 : x_s0;
        for (int i = t; i < (P + 2) * (C / 8) * 2; i += 512) {
            const int j = i / ((C / 8) * 2);
            const int rem = i % ((C / 8) * 2);
            const int col = (rem & 1) ? (W + 1) : 0;
            const int cg = (rem >> 1) * 8;
            u32x4 z = {0u, 0u, 0u, 0u};
            *reinterpret_cast<u32x4*>(
                &x_s[(j * LP + col) * CS + cg]) = z;
        }
    }

    const int lines_per_img = H / P;
    const long chunks_total = (long)N * lines_per_img;

    MmRegs regs;
    // Prologue: stage chunk 0 into buffer 0; issue chunk 1's loads.
    if (sp < chunks_total) {
        const int n0 = (int)(sp / lines_per_img);
        const int h00 = (int)(sp % lines_per_img) * P;
        mm_issue(x, n0, h00, H, W, C, 0, P, t, regs);
        mm_write(x_s0, W, C, CS, LP, P, t, regs);
    }
    __syncthreads();
    // Both paths prefetch chunk sp+spread now: with dbuf it lands in
    // the other LDS buffer; without, it stays in regs through the
    // contraction of chunk sp (register double-buffering — the 160 KB
    // LDS can't hold two x windows at C=128, and an un-prefetched
    // single-buffer path exposes the full global-load latency between
    // chunks: layer2 measured 259 us vs layer1's pipelined 174).
    if (sp + spread < chunks_total) {
        const long qn = sp + spread;
        mm_issue(x, (int)(qn / lines_per_img),
                 (int)(qn % lines_per_img) * P, H, W, C, 0, P, t, regs);
    }

    int cur = 0;
    for (long q = sp; q < chunks_total; q += spread) {
        const int n = (int)(q / lines_per_img);
        const int h0 = (int)(q % lines_per_img) * P;
        const short* x_s = (dbuf && cur) ? x_s1 : x_s0;

        if (!dbuf && q != sp) {
            // regs already hold chunk q (prefetched during the previous
            // contraction); commit it to LDS, then start chunk q+spread's
            // loads so they fly behind this chunk's MFMA work.
            __syncthreads();   // all waves done with x_s (y-store bounce)
            mm_write(x_s0, W, C, CS, LP, P, t, regs);
            __syncthreads();
            if (q + spread < chunks_total) {
                const long qn = q + spread;
                mm_issue(x, (int)(qn / lines_per_img),
                         (int)(qn % lines_per_img) * P, H, W, C, 0, P, t,
                         regs);
            }
        }

        // ---- contraction: 9 taps x C/32 channel chunks ----
        // acc split by tau parity: a fragment's 9 taps form one serial
        // MFMA dependency chain per cc iteration otherwise (KT=32 has
        // only 2 (mf) chains per wave -- suspected cause of layer2's
        // 255 us, ROADMAP item 1).  Parity doubles the chains.
        f32x4 acc[2][2][2];
        #pragma unroll
        for (int mf = 0; mf < 2; ++mf)
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf)
                #pragma unroll
                for (int tp = 0; tp < 2; ++tp)
                    #pragma unroll
                    for (int r = 0; r < 4; ++r)
                        acc[mf][nf][tp][r] = 0.f;

        // cc outer (runtime trip count, stays rolled); tau inner and
        // fully unrolled: a rolled tau loop compiled to 4 MFMAs each
        // preceded by s_waitcnt lgkmcnt(0) — LDS reads and MFMA issue
        // fully serialized.  Unrolled, the body exposes 36 MFMAs + 36
        // independent b128 reads of ILP per cc iteration.
        for (int cc = 0; cc < C / 32; ++cc) {
            #pragma unroll
            for (int tau = 0; tau < 9; ++tau) {
                const int dh = tau / 3;
                const int dw = tau % 3;
                bf16x8 bfrag[2];
                #pragma unroll
                for (int nf = 0; nf < 2; ++nf) {
                    if (nf < NF) {
                        const int ko = wk * KH + nf * 16 + row16;
                        bfrag[nf] = *reinterpret_cast<const bf16x8*>(
                            &w_lds[(ko * 9 + tau) * WS + cc * 32
                                   + slot8]);
                    }
                }
                #pragma unroll
                for (int mf = 0; mf < 2; ++mf) {
                    const int p = wp * 32 + mf * 16 + row16;
                    const int li = p / W;
                    const int ww = p % W;
                    const int qrow = (li + dh) * LP + ww + dw;
                    const bf16x8 afrag =
                        *reinterpret_cast<const bf16x8*>(
                            &x_s[qrow * CS + cc * 32 + slot8]);
                    #pragma unroll
                    for (int nf = 0; nf < 2; ++nf)
                        if (nf < NF)
                            acc[mf][nf][tau & 1] =
                                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                    afrag, bfrag[nf], acc[mf][nf][tau & 1],
                                    0, 0, 0);
                }
            }
        }

        // ---- write the y tile: bounce through the (now free) current
        // x_s buffer so global stores are coalesced 16B lines instead
        // of 16 scattered 2B stores per lane (store-issue bound:
        // phase probe measured them at 47 us of the 187 us call). ----
#ifdef MM_PROBE_NOSTORE
        if (ybase_guard(acc[0][0][0])) continue;
#endif
        const size_t ybase = ((size_t)n * H + h0) * W;  // chunk pixel 0
        short* ystage = const_cast<short*>(x_s);        // [128][KT]
        __syncthreads();  // all waves done reading x_s
        #pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf) {
                if (nf >= NF) continue;
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int p = wp * 32 + mf * 16 + (lane >> 4) * 4
                        + r;
                    const int ko = wk * KH + nf * 16 + (lane & 15);
                    ystage[p * KT + ko] =
                        (short)f2b(acc[mf][nf][0][r]
                                   + acc[mf][nf][1][r]);
                }
            }
        }
        __syncthreads();
        {
            const int pieces = CP * KT / 8;   // 16B pieces
            for (int i = t; i < pieces; i += 512) {
                const int p = i / (KT / 8);
                const int kg = (i % (KT / 8)) * 8;
                V16 v;
                v.u4 = *reinterpret_cast<const u32x4*>(
                    &ystage[p * KT + kg]);
                *reinterpret_cast<u32x4*>(
                    const_cast<ushort_t*>(y + (ybase + p) * K
                                          + kt * KT + kg)) = v.u4;
            }
        }

        // ---- pipelined: stage q+1 into the other buffer, issue q+2 --
        if (dbuf) {
            if (q + spread < chunks_total) {
                short* nxt = cur ? x_s0 : x_s1;
                mm_write(nxt, W, C, CS, LP, P, t, regs);
                if (q + 2 * spread < chunks_total) {
                    const long q2 = q + 2 * spread;
                    mm_issue(x, (int)(q2 / lines_per_img),
                             (int)(q2 % lines_per_img) * P, H, W, C, 0,
                             P, t, regs);
                }
            }
            __syncthreads();
            cur ^= 1;
        }
    }
}

extern "C" int conv3x3_mm_supported(int H, int W, int C, int K) {
    if (C % 32 || C > 128) return 0;
    if (W != 16 && W != 32) return 0;
    int P = (W == 32) ? 4 : 8;
    if (H % P) return 0;
    int KT = (C <= 64) ? 64 : 32;
    if (K % KT) return 0;
    return 1;
}

extern "C" void launch_conv3x3_mm(
        const ushort_t* x, const ushort_t* w, ushort_t* y,
        int N, int H, int W, int C, int K, hipStream_t s) {
    const int P = (W == 32) ? 4 : 8;
    const int KT = (C <= 64) ? 64 : 32;
    const int CS = C + 16;
    const int WS = C + 8;
    const int QR = (P + 2) * (W + 2);
    // double-buffer the x window when it fits the 160 KB LDS
    size_t lds_bytes = ((size_t)KT * 9 * WS + 2 * (size_t)QR * CS) * 2;
    int dbuf = 1;
    if (lds_bytes > 160 * 1024) {
        dbuf = 0;
        lds_bytes = ((size_t)KT * 9 * WS + (size_t)QR * CS) * 2;
    }
    static int attr_set = 0;
    if (!attr_set) {
        hipFuncSetAttribute(
            reinterpret_cast<const void*>(&k_conv3x3_mm),
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = 1;
    }
    const long chunks = (long)N * (H / P);
    const int nk = K / KT;
    // ~112-127 KB dynamic LDS -> 1 block/CU: size the grid to exactly
    // one residency round so the weight tile is staged once per CU.
    long spread = 256 / nk;
    if (spread > chunks) spread = chunks;
    if (spread < 1) spread = 1;
    hipLaunchKernelGGL(k_conv3x3_mm, dim3((unsigned)(nk * spread)),
                       dim3(512), lds_bytes, s, x, w, y, N, H, W, C, K,
                       P, KT, (int)spread, dbuf);
}

// =====================================================================
// MFMA 3x3 STRIDE-2 backward-data (NHWC bf16), polyphase decomposition.
//
// dx[n,hi,wi,c] = sum_{k,dh,dw : hi=2ho+dh-1, wi=2wo+dw-1} dy[n,ho,wo,k]
//                 * w[k,dh,dw,c]
//
// Profiling (profiles/bench_r05_kernel_stats.csv) shows the stride-2
// bwd-data falling to a CK kernel at ~1035 us/call (~75 GF/s) -- the
// single largest per-call conv entry of the ResNet step.  Splitting dx
// by pixel parity (a,b) = (hi%2, wi%2) turns the strided scatter into
// four dense stride-1 mini-convs over dy with disjoint tap sets:
//
//   phase (a,b): taps dh in (a ? {0,2} : {1}) x dw in (b ? {0,2} : {1})
//   ho = h2 + (dh==0), wo = w2 + (dw==0)      (hi=2*h2+a, wi=2*w2+b)
//
// (1 + 2 + 2 + 4 taps = the 9 taps, each used exactly once.)
// Block layout (conv_mm skeleton: natural-layout staging, all fragment
// reads aligned b128, register-prefetch pipeline, y-store bounce):
//   - a chunk is P2 phase-rows = 2*P2 full dx lines of one image; the
//     dy window is P2+1 lines (bottom halo only: tap offsets are >= 0),
//     pixel-major [line][wo (+8 zero pad cols)][K+16 stride],
//   - w_lds holds the TRANSPOSED weight tile wt[c][tau][k] (contiguous
//     in the contraction dim k; prepared host-side as a [C][3][3][K]
//     tensor -- cheap: weights are tiny),
//   - waves: 4 pixel-quarters x (CT/16 = 2) c-frags; each quarter owns
//     one fragment of phase pair {(0,0),(1,1)} or {(0,1),(1,0)} so the
//     per-wave tap count is 5 vs 4 (1+4 / 2+2) instead of the 4x
//     imbalance of phase-per-quarter,
//   - the chunk's 4 phases together tile 2*P2 COMPLETE dx lines, so the
//     bounce-staged store writes plain contiguous b128 lines.
// Constraints (v1): Wo == 16 (so a 16-pixel fragment row is one phase
// line), Ho % P2 == 0, K % 32 == 0, C % 32 == 0, CT = 32.
// =====================================================================

#define S2_P2 2
#define S2_CT 32

struct S2Regs {
    V16 v[2];
};

__device__ __forceinline__ void s2_issue(
        const ushort_t* __restrict__ dy, int n, int r0, int Ho, int Wo,
        int K, int t, S2Regs& r) {
    const int pieces = (S2_P2 + 1) * Wo * (K / 8);
    #pragma unroll
    for (int it = 0; it < 2; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (Wo * (K / 8));
            const int rem = i % (Wo * (K / 8));
            const int wo = rem / (K / 8);
            const int kg = (rem % (K / 8)) * 8;
            const int ho = r0 + j;
            if (ho >= Ho) {
                r.v[it].u4 = u32x4{0u, 0u, 0u, 0u};
            } else {
                r.v[it].u4 = *reinterpret_cast<const u32x4*>(
                    dy + (((size_t)n * Ho + ho) * Wo + wo) * K + kg);
            }
        }
    }
}

__device__ __forceinline__ void s2_write(
        short* __restrict__ dy_s, int Wo, int K, int KS, int LW, int t,
        S2Regs& r) {
    const int pieces = (S2_P2 + 1) * Wo * (K / 8);
    #pragma unroll
    for (int it = 0; it < 2; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (Wo * (K / 8));
            const int rem = i % (Wo * (K / 8));
            const int wo = rem / (K / 8);
            const int kg = (rem % (K / 8)) * 8;
            *reinterpret_cast<u32x4*>(
                &dy_s[(j * LW + wo) * KS + kg]) = r.v[it].u4;
        }
    }
    // Refresh the zero pad columns every staging: the dx bounce store
    // reuses this buffer and clobbers them.
    const int zp = (S2_P2 + 1) * 8 * (K / 8);
    for (int i = t; i < zp; i += 512) {
        const int j = i / (8 * (K / 8));
        const int rem = i % (8 * (K / 8));
        const int pc = rem / (K / 8);
        const int kg = (rem % (K / 8)) * 8;
        u32x4 z = {0u, 0u, 0u, 0u};
        *reinterpret_cast<u32x4*>(
            &dy_s[(j * LW + Wo + pc) * KS + kg]) = z;
    }
}

// Contraction for wave-quarter Q.  Fragment f's phase: quarters 0..1
// own {(0,0),(1,1)}, quarters 2..3 own {(0,1),(1,0)}; h2 = Q&1.
// KC = K/32 as a compile-time trip count (a runtime bound kept the loop
// rolled -- conv_mm lesson); cc-parity-split accumulators double the
// independent MFMA chains per wave (2 -> 4) to cover MFMA latency.
template <int Q, int KC>
__device__ __forceinline__ void s2_contract(
        const short* __restrict__ dy_s, const short* __restrict__ w_lds,
        int KS, int KS2, int LW, int wn, int row16, int slot8,
        f32x4 (&acc)[2][2]) {
    const int h2 = Q & 1;
    const int w2 = row16;            // Wo == 16: fragment row == w2
    #pragma unroll
    for (int cc = 0; cc < KC; ++cc) {
        #pragma unroll
        for (int f = 0; f < 2; ++f) {
            constexpr int PH0 = (Q < 2) ? 0 : 1;   // f == 0 phase
            constexpr int PH1 = (Q < 2) ? 3 : 2;   // f == 1 phase
            const int ph = f ? PH1 : PH0;
            const int a = ph >> 1, b = ph & 1;
            #pragma unroll
            for (int dh = 0; dh < 3; ++dh) {
                if (a ? (dh == 1) : (dh != 1)) continue;
                #pragma unroll
                for (int dw = 0; dw < 3; ++dw) {
                    if (b ? (dw == 1) : (dw != 1)) continue;
                    const int j = h2 + (dh == 0 ? 1 : 0);
                    const int col = w2 + (dw == 0 ? 1 : 0);
                    const bf16x8 afrag =
                        *reinterpret_cast<const bf16x8*>(
                            &dy_s[(j * LW + col) * KS + cc * 32
                                  + slot8]);
                    const int clocal = wn * 16 + row16;
                    const int tau = dh * 3 + dw;
                    const bf16x8 bfrag =
                        *reinterpret_cast<const bf16x8*>(
                            &w_lds[((clocal * 9) + tau) * KS2
                                   + cc * 32 + slot8]);
                    acc[f][cc & 1] =
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            afrag, bfrag, acc[f][cc & 1], 0, 0, 0);
                }
            }
        }
    }
}

template <int Q>
__device__ __forceinline__ void s2_contract_k(
        const short* __restrict__ dy_s, const short* __restrict__ w_lds,
        int K, int KS, int KS2, int LW, int wn, int row16, int slot8,
        f32x4 (&acc)[2][2]) {
    switch (K / 32) {
    case 2: s2_contract<Q, 2>(dy_s, w_lds, KS, KS2, LW, wn, row16,
                              slot8, acc); break;
    case 4: s2_contract<Q, 4>(dy_s, w_lds, KS, KS2, LW, wn, row16,
                              slot8, acc); break;
    case 8: s2_contract<Q, 8>(dy_s, w_lds, KS, KS2, LW, wn, row16,
                              slot8, acc); break;
    default: break;  // unreachable: supported() gates K to {64,128,256}
    }
}

extern "C" __global__ __launch_bounds__(512, 2) void k_conv3x3_s2_bwd(
        const ushort_t* __restrict__ dy, const ushort_t* __restrict__ wt,
        ushort_t* __restrict__ dx, int N, int Ho, int Wo, int K, int C,
        int spread, int dbuf) {
    extern __shared__ short lds[];
    const int KS = K + 16;           // dy_s row stride (zero conflicts)
    const int KS2 = K + 8;           // w_lds row stride
    const int LW = Wo + 8;           // dy line width incl. zero pad cols
    const int WIN = (S2_P2 + 1) * LW;
    short* w_lds = lds;              // [S2_CT * 9][KS2]
    short* dy_s0 = lds + S2_CT * 9 * KS2;
    short* dy_s1 = dy_s0 + (dbuf ? WIN * KS : 0);

    const int Hi = 2 * Ho, Wi = 2 * Wo;
    const int nct = C / S2_CT;
    const int ct = blockIdx.x / spread;
    const int sp = blockIdx.x % spread;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;
    const int wq = wid >> 1;         // pixel quarter (phase pair + h2)
    const int wn = wid & 1;          // c fragment within the CT tile
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;

    // ---- stage the transposed weight tile wt[c][tau][k] once ----
    {
        const int pieces = S2_CT * 9 * (K / 8);
        const ushort_t* wg = wt + (size_t)ct * S2_CT * 9 * K;
        for (int i = t; i < pieces; i += 512) {
            const int row = i / (K / 8);          // clocal * 9 + tau
            const int kg = (i % (K / 8)) * 8;
            *reinterpret_cast<u32x4*>(&w_lds[row * KS2 + kg]) =
                *reinterpret_cast<const u32x4*>(
                    wg + (size_t)row * K + kg);
        }
    }
    // (pad columns are zeroed inside s2_write on every staging)
    const int rows_per_img = Ho / S2_P2;
    const long chunks_total = (long)N * rows_per_img;

    S2Regs regs;
    if (sp < chunks_total) {
        s2_issue(dy, (int)(sp / rows_per_img),
                 (int)(sp % rows_per_img) * S2_P2, Ho, Wo, K, t, regs);
        s2_write(dy_s0, Wo, K, KS, LW, t, regs);
    }
    __syncthreads();
    if (sp + spread < chunks_total) {
        const long qn = sp + spread;
        s2_issue(dy, (int)(qn / rows_per_img),
                 (int)(qn % rows_per_img) * S2_P2, Ho, Wo, K, t, regs);
    }

    int cur = 0;
    for (long q = sp; q < chunks_total; q += spread) {
        const int n = (int)(q / rows_per_img);
        const int r0 = (int)(q % rows_per_img) * S2_P2;
        short* dy_s = (dbuf && cur) ? dy_s1 : dy_s0;

        if (!dbuf && q != sp) {
            __syncthreads();
            s2_write(dy_s0, Wo, K, KS, LW, t, regs);
            __syncthreads();
            if (q + spread < chunks_total) {
                const long qn = q + spread;
                s2_issue(dy, (int)(qn / rows_per_img),
                         (int)(qn % rows_per_img) * S2_P2, Ho, Wo, K, t,
                         regs);
            }
        }

        f32x4 acc[2][2];
        #pragma unroll
        for (int f = 0; f < 2; ++f)
            #pragma unroll
            for (int pp = 0; pp < 2; ++pp)
                #pragma unroll
                for (int r = 0; r < 4; ++r)
                    acc[f][pp][r] = 0.f;

        switch (wq) {
        case 0: s2_contract_k<0>(dy_s, w_lds, K, KS, KS2, LW, wn, row16,
                                 slot8, acc); break;
        case 1: s2_contract_k<1>(dy_s, w_lds, K, KS, KS2, LW, wn, row16,
                                 slot8, acc); break;
        case 2: s2_contract_k<2>(dy_s, w_lds, K, KS, KS2, LW, wn, row16,
                                 slot8, acc); break;
        default: s2_contract_k<3>(dy_s, w_lds, K, KS, KS2, LW, wn,
                                  row16, slot8, acc); break;
        }

        // ---- bounce the dx tile through the (now free) dy_s buffer so
        // global stores go out as whole 16B lines ----
        ushort_t* ystage = reinterpret_cast<ushort_t*>(dy_s);
        __syncthreads();             // all waves done reading dy_s
        #pragma unroll
        for (int f = 0; f < 2; ++f) {
            const int ph = (wq < 2) ? (f ? 3 : 0) : (f ? 2 : 1);
            const int h2 = wq & 1;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int w2 = (lane >> 4) * 4 + r;
                const int c = wn * 16 + (lane & 15);
                const int s = (ph * S2_P2 + h2) * Wo + w2;
                ystage[s * S2_CT + c] =
                    f2b(acc[f][0][r] + acc[f][1][r]);
            }
        }
        __syncthreads();
        {
            const int pieces = 2 * S2_P2 * Wi * (S2_CT / 8);
            const size_t base = ((size_t)n * Hi + 2 * r0) * Wi;
            for (int i = t; i < pieces; i += 512) {
                const int hl = i / (Wi * (S2_CT / 8));
                const int rem = i % (Wi * (S2_CT / 8));
                const int wi = rem / (S2_CT / 8);
                const int cg = (rem % (S2_CT / 8)) * 8;
                const int ph = ((hl & 1) << 1) | (wi & 1);
                const int s = (ph * S2_P2 + (hl >> 1)) * Wo + (wi >> 1);
                *reinterpret_cast<u32x4*>(
                    dx + (base + (size_t)hl * Wi + wi) * C
                    + ct * S2_CT + cg) =
                    *reinterpret_cast<const u32x4*>(
                        &ystage[s * S2_CT + cg]);
            }
        }

        if (dbuf) {
            if (q + spread < chunks_total) {
                short* nxt = cur ? dy_s0 : dy_s1;
                s2_write(nxt, Wo, K, KS, LW, t, regs);
                if (q + 2 * spread < chunks_total) {
                    const long q2 = q + 2 * spread;
                    s2_issue(dy, (int)(q2 / rows_per_img),
                             (int)(q2 % rows_per_img) * S2_P2, Ho, Wo, K,
                             t, regs);
                }
            }
            __syncthreads();
            cur ^= 1;
        }
    }
}

// ---- Wo == 8 variant: one 16-pixel fragment is a WHOLE phase image
// of the chunk (2 rows x 8), so waves split the K contraction in
// quarters instead of the c tile, and the four partial dx tiles merge
// through an f32 LDS stage (ds atomics) before the bf16 bounce store.
// CT = 16 keeps the K=256 weight tile inside LDS (76 KB + 2x26 KB
// double-buffered dy windows = 128 KB).

#define S2W8_CT 16

template <int G, int KQ, int KC>
__device__ __forceinline__ void s2w8_contract(
        const short* __restrict__ dy_s, const short* __restrict__ w_lds,
        int KS, int KS2, int LW, int row16, int slot8,
        f32x4 (&acc)[2][2]) {
    const int h2 = row16 >> 3;
    const int w2 = row16 & 7;
    #pragma unroll
    for (int cc = 0; cc < KC; ++cc) {
        const int kbase = KQ * (KC * 32) + cc * 32;
        #pragma unroll
        for (int f = 0; f < 2; ++f) {
            constexpr int PH0 = (G == 0) ? 0 : 1;
            constexpr int PH1 = (G == 0) ? 3 : 2;
            const int ph = f ? PH1 : PH0;
            const int a = ph >> 1, b = ph & 1;
            #pragma unroll
            for (int dh = 0; dh < 3; ++dh) {
                if (a ? (dh == 1) : (dh != 1)) continue;
                #pragma unroll
                for (int dw = 0; dw < 3; ++dw) {
                    if (b ? (dw == 1) : (dw != 1)) continue;
                    const int j = h2 + (dh == 0 ? 1 : 0);
                    const int col = w2 + (dw == 0 ? 1 : 0);
                    const bf16x8 afrag =
                        *reinterpret_cast<const bf16x8*>(
                            &dy_s[(j * LW + col) * KS + kbase + slot8]);
                    const int tau = dh * 3 + dw;
                    const bf16x8 bfrag =
                        *reinterpret_cast<const bf16x8*>(
                            &w_lds[(row16 * 9 + tau) * KS2 + kbase
                                   + slot8]);
                    acc[f][cc & 1] =
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            afrag, bfrag, acc[f][cc & 1], 0, 0, 0);
                }
            }
        }
    }
}

template <int KC>
__device__ __forceinline__ void s2w8_contract_g(
        int wg, const short* __restrict__ dy_s,
        const short* __restrict__ w_lds, int KS, int KS2, int LW,
        int row16, int slot8, f32x4 (&acc)[2][2]) {
    switch (wg) {
    case 0: s2w8_contract<0, 0, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    case 1: s2w8_contract<0, 1, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    case 2: s2w8_contract<0, 2, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    case 3: s2w8_contract<0, 3, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    case 4: s2w8_contract<1, 0, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    case 5: s2w8_contract<1, 1, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    case 6: s2w8_contract<1, 2, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                    slot8, acc); break;
    default: s2w8_contract<1, 3, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    }
}

extern "C" __global__ __launch_bounds__(512, 2) void k_conv3x3_s2_bwd_w8(
        const ushort_t* __restrict__ dy, const ushort_t* __restrict__ wt,
        ushort_t* __restrict__ dx, int N, int Ho, int K, int C,
        int spread, int dbuf) {
    extern __shared__ short lds[];
    const int Wo = 8;
    const int KS = K + 16;
    const int KS2 = K + 8;
    const int LW = Wo + 8;
    const int WIN = (S2_P2 + 1) * LW;
    short* w_lds = lds;              // [S2W8_CT * 9][KS2]
    short* dy_s0 = lds + S2W8_CT * 9 * KS2;
    short* dy_s1 = dy_s0 + (dbuf ? WIN * KS : 0);

    const int Hi = 2 * Ho, Wi = 2 * Wo;
    const int ct = blockIdx.x / spread;
    const int sp = blockIdx.x % spread;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;          // = 4*group + k-quarter
    const int wg2 = wid >> 2;        // phase-pair group
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;

    // CP2 = 4 phases * P2 * Wo = 64 chunk pixels
    const int CP2 = 4 * S2_P2 * Wo;

    {
        const int pieces = S2W8_CT * 9 * (K / 8);
        const ushort_t* wg = wt + (size_t)ct * S2W8_CT * 9 * K;
        for (int i = t; i < pieces; i += 512) {
            const int row = i / (K / 8);
            const int kg = (i % (K / 8)) * 8;
            *reinterpret_cast<u32x4*>(&w_lds[row * KS2 + kg]) =
                *reinterpret_cast<const u32x4*>(
                    wg + (size_t)row * K + kg);
        }
    }

    const int rows_per_img = Ho / S2_P2;
    const long chunks_total = (long)N * rows_per_img;

    S2Regs regs;
    if (sp < chunks_total) {
        s2_issue(dy, (int)(sp / rows_per_img),
                 (int)(sp % rows_per_img) * S2_P2, Ho, Wo, K, t, regs);
        s2_write(dy_s0, Wo, K, KS, LW, t, regs);
    }
    __syncthreads();
    if (sp + spread < chunks_total) {
        const long qn = sp + spread;
        s2_issue(dy, (int)(qn / rows_per_img),
                 (int)(qn % rows_per_img) * S2_P2, Ho, Wo, K, t, regs);
    }

    int cur = 0;
    for (long q = sp; q < chunks_total; q += spread) {
        const int n = (int)(q / rows_per_img);
        const int r0 = (int)(q % rows_per_img) * S2_P2;
        short* dy_s = (dbuf && cur) ? dy_s1 : dy_s0;

        if (!dbuf && q != sp) {
            __syncthreads();
            s2_write(dy_s0, Wo, K, KS, LW, t, regs);
            __syncthreads();
            if (q + spread < chunks_total) {
                const long qn = q + spread;
                s2_issue(dy, (int)(qn / rows_per_img),
                         (int)(qn % rows_per_img) * S2_P2, Ho, Wo, K, t,
                         regs);
            }
        }

        f32x4 acc[2][2];
        #pragma unroll
        for (int f = 0; f < 2; ++f)
            #pragma unroll
            for (int pp = 0; pp < 2; ++pp)
                #pragma unroll
                for (int r = 0; r < 4; ++r)
                    acc[f][pp][r] = 0.f;

        if (K / 32 == 8)
            s2w8_contract_g<2>(wid & 7, dy_s, w_lds, KS, KS2, LW,
                               row16, slot8, acc);
        else
            s2w8_contract_g<1>(wid & 7, dy_s, w_lds, KS, KS2, LW,
                               row16, slot8, acc);

        // ---- merge the 4 k-quarter partials in an f32 LDS stage ----
        float* fstage = reinterpret_cast<float*>(dy_s);  // [64][CT]
        __syncthreads();             // all waves done reading dy_s
        for (int i = t; i < CP2 * S2W8_CT; i += 512)
            fstage[i] = 0.f;
        __syncthreads();
        #pragma unroll
        for (int f = 0; f < 2; ++f) {
            const int ph = (wg2 == 0) ? (f ? 3 : 0) : (f ? 2 : 1);
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = (lane >> 4) * 4 + r;
                const int h2 = m >> 3, w2 = m & 7;
                const int c = lane & 15;
                const int s = (ph * S2_P2 + h2) * Wo + w2;
                atomicAdd(&fstage[s * S2W8_CT + c],
                          acc[f][0][r] + acc[f][1][r]);
            }
        }
        __syncthreads();
        {
            // convert + interleaved contiguous store (8 c per piece)
            const int pieces = 2 * S2_P2 * Wi * (S2W8_CT / 8);
            const size_t base = ((size_t)n * Hi + 2 * r0) * Wi;
            for (int i = t; i < pieces; i += 512) {
                const int hl = i / (Wi * (S2W8_CT / 8));
                const int rem = i % (Wi * (S2W8_CT / 8));
                const int wi = rem / (S2W8_CT / 8);
                const int cg = (rem % (S2W8_CT / 8)) * 8;
                const int ph = ((hl & 1) << 1) | (wi & 1);
                const int s = (ph * S2_P2 + (hl >> 1)) * Wo
                    + (wi >> 1);
                ushort_t out[8];
                #pragma unroll
                for (int jj = 0; jj < 8; ++jj)
                    out[jj] = f2b(fstage[s * S2W8_CT + cg + jj]);
                *reinterpret_cast<u32x4*>(
                    dx + (base + (size_t)hl * Wi + wi) * C
                    + ct * S2W8_CT + cg) =
                    *reinterpret_cast<const u32x4*>(out);
            }
        }

        if (dbuf) {
            if (q + spread < chunks_total) {
                short* nxt = cur ? dy_s0 : dy_s1;
                s2_write(nxt, Wo, K, KS, LW, t, regs);
                if (q + 2 * spread < chunks_total) {
                    const long q2 = q + 2 * spread;
                    s2_issue(dy, (int)(q2 / rows_per_img),
                             (int)(q2 % rows_per_img) * S2_P2, Ho, Wo,
                             K, t, regs);
                }
            }
            __syncthreads();
            cur ^= 1;
        }
    }
}

// =====================================================================
// MFMA 3x3 STRIDE-2 FORWARD (NHWC bf16), experimental.
//
//   y[n,ho,wo,k] = sum_{dh,dw,c} x[n,2ho+dh-1,2wo+dw-1,c] * w[k,tau,c]
//
// Unlike backward-data, the forward is a pure gather: with pixel-major
// LDS staging each tap is just a per-lane address offset.  The staging
// splits every input row into even|odd COLUMN sections ("E|z|O", row
// stride LS2 = 2*Wo+2, the z column holds the wo=0 left-pad zero), so
// a fragment's 16 lanes step by ONE column slot = CS2 = C+8 shorts
// (36 dwords at C=64: 16 distinct banks) instead of two raw columns
// (2*(C+16) = 80 dwords: 4 banks, 4-way conflicts).  Tap decode:
//   dw=1 -> E[wo];  dw=0 -> O[wo-1];  dw=2 -> O[wo]
//   local row L = 2*(ho-r0) + dh  (L=0 is input row 2*r0-1; top halo
//   only -- the bottom tap row 2*(r0+P2-1)+1 is always in range).
// Weight tile w_lds[ko][tau][ci] and the wave layout mirror conv_mm;
// chunk = P2F y-rows (CP = P2F*Wo = 64 pixels: 4 one-fragment pixel
// quarters x 2 ko-halves of the KT=64 tile).  Single-buffered x window
// with register prefetch (the w tile + two windows exceed 160 KB).
// Constraints: Wo == 16 (fragment row == one y line), Ho % P2F == 0,
// C % 32 == 0, C <= 128, K % 64 == 0.
// =====================================================================

#define S2F_P2 4

struct S2FRegs {
    V16 v[5];
};

__device__ __forceinline__ void s2f_issue(
        const ushort_t* __restrict__ x, int n, int r0, int H, int W,
        int C, int t, S2FRegs& r) {
    // stage input rows 2*r0-1 .. 2*r0+2*P2F-1 (2*P2F+1 rows)
    const int pieces = (2 * S2F_P2 + 1) * W * (C / 8);
    #pragma unroll
    for (int it = 0; it < 5; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (W * (C / 8));
            const int rem = i % (W * (C / 8));
            const int wcol = rem / (C / 8);
            const int cg = (rem % (C / 8)) * 8;
            const int h = 2 * r0 - 1 + j;
            if (h < 0 || h >= H) {
                r.v[it].u4 = u32x4{0u, 0u, 0u, 0u};
            } else {
                r.v[it].u4 = *reinterpret_cast<const u32x4*>(
                    x + (((size_t)n * H + h) * W + wcol) * C + cg);
            }
        }
    }
}

__device__ __forceinline__ void s2f_write(
        short* __restrict__ x_s, int W, int C, int CS2, int LS2, int t,
        S2FRegs& r) {
    const int Wo = W / 2;
    const int pieces = (2 * S2F_P2 + 1) * W * (C / 8);
    #pragma unroll
    for (int it = 0; it < 5; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (W * (C / 8));
            const int rem = i % (W * (C / 8));
            const int wcol = rem / (C / 8);
            const int cg = (rem % (C / 8)) * 8;
            // E section: slots 0..Wo-1; zero col at Wo; O: Wo+1..2Wo
            const int slot = (wcol & 1) ? (Wo + 1 + (wcol >> 1))
                                        : (wcol >> 1);
            *reinterpret_cast<u32x4*>(
                &x_s[(j * LS2 + slot) * CS2 + cg]) = r.v[it].u4;
        }
    }
    // left-pad zero column (O[-1]) of every staged row
    const int zp = (2 * S2F_P2 + 1) * (C / 8);
    for (int i = t; i < zp; i += 512) {
        const int j = i / (C / 8);
        const int cg = (i % (C / 8)) * 8;
        u32x4 z = {0u, 0u, 0u, 0u};
        *reinterpret_cast<u32x4*>(
            &x_s[(j * LS2 + Wo) * CS2 + cg]) = z;
    }
}

template <int CBLK>
__device__ __forceinline__ void s2f_contract(
        const short* __restrict__ x_s, const short* __restrict__ w_lds,
        int CS2, int WS, int LS2, int Wo, int wq, int wk, int row16,
        int slot8, f32x4 (&acc)[2]) {
    const int ho_loc = wq;           // quarter = one y line (Wo == 16)
    const int wo = row16;
    #pragma unroll
    for (int cc = 0; cc < CBLK; ++cc) {
        #pragma unroll
        for (int dh = 0; dh < 3; ++dh) {
            #pragma unroll
            for (int dw = 0; dw < 3; ++dw) {
                const int L = 2 * ho_loc + dh;
                const int slot = (dw == 1) ? wo
                    : (Wo + 1 + wo - (dw == 0 ? 1 : 0));
                const bf16x8 afrag =
                    *reinterpret_cast<const bf16x8*>(
                        &x_s[(L * LS2 + slot) * CS2 + cc * 32
                             + slot8]);
                const int tau = dh * 3 + dw;
                #pragma unroll
                for (int nf = 0; nf < 2; ++nf) {
                    const int ko = wk * 32 + nf * 16 + row16;
                    const bf16x8 bfrag =
                        *reinterpret_cast<const bf16x8*>(
                            &w_lds[(ko * 9 + tau) * WS + cc * 32
                                   + slot8]);
                    acc[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        afrag, bfrag, acc[nf], 0, 0, 0);
                }
            }
        }
    }
}

extern "C" __global__ __launch_bounds__(512, 2) void k_conv3x3_s2_fwd(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ w,
        ushort_t* __restrict__ y, int N, int H, int W, int C, int K,
        int spread) {
    extern __shared__ short lds[];
    const int KT = 64;
    const int Wo = W / 2, Ho = H / 2;
    const int CS2 = C + 8;
    const int WS = C + 8;
    const int LS2 = 2 * Wo + 2;
    short* w_lds = lds;              // [KT * 9][WS]
    short* x_s = lds + KT * 9 * WS;  // [(2*P2F+1) rows][LS2][CS2]

    const int nk = K / KT;
    const int kt = blockIdx.x / spread;
    const int sp = blockIdx.x % spread;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;
    const int wq = wid >> 1;         // y-line quarter of the chunk
    const int wk = wid & 1;          // ko half of the KT tile
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;
    const int CP = S2F_P2 * Wo;      // 64 chunk pixels

    {
        const int pieces = KT * 9 * (C / 8);
        const ushort_t* wg = w + (size_t)kt * KT * 9 * C;
        for (int i = t; i < pieces; i += 512) {
            const int row = i / (C / 8);
            const int cg = (i % (C / 8)) * 8;
            *reinterpret_cast<u32x4*>(&w_lds[row * WS + cg]) =
                *reinterpret_cast<const u32x4*>(
                    wg + (size_t)row * C + cg);
        }
    }

    const int rows_per_img = Ho / S2F_P2;
    const long chunks_total = (long)N * rows_per_img;

    S2FRegs regs;
    if (sp < chunks_total) {
        s2f_issue(x, (int)(sp / rows_per_img),
                  (int)(sp % rows_per_img) * S2F_P2, H, W, C, t, regs);
        s2f_write(x_s, W, C, CS2, LS2, t, regs);
    }
    __syncthreads();
    if (sp + spread < chunks_total) {
        const long qn = sp + spread;
        s2f_issue(x, (int)(qn / rows_per_img),
                  (int)(qn % rows_per_img) * S2F_P2, H, W, C, t, regs);
    }

    for (long q = sp; q < chunks_total; q += spread) {
        const int n = (int)(q / rows_per_img);
        const int r0 = (int)(q % rows_per_img) * S2F_P2;

        if (q != sp) {
            __syncthreads();         // y bounce of q-1 fully read
            s2f_write(x_s, W, C, CS2, LS2, t, regs);
            __syncthreads();
            if (q + spread < chunks_total) {
                const long qn = q + spread;
                s2f_issue(x, (int)(qn / rows_per_img),
                          (int)(qn % rows_per_img) * S2F_P2, H, W, C, t,
                          regs);
            }
        }

        f32x4 acc[2];
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf)
            #pragma unroll
            for (int r = 0; r < 4; ++r)
                acc[nf][r] = 0.f;

        switch (C / 32) {
        case 2: s2f_contract<2>(x_s, w_lds, CS2, WS, LS2, Wo, wq, wk,
                                row16, slot8, acc); break;
        case 3: s2f_contract<3>(x_s, w_lds, CS2, WS, LS2, Wo, wq, wk,
                                row16, slot8, acc); break;
        default: s2f_contract<4>(x_s, w_lds, CS2, WS, LS2, Wo, wq, wk,
                                 row16, slot8, acc); break;
        }

        // bounce y tile through x_s: [64 px][KT] bf16 = 8 KB
        ushort_t* ystage = reinterpret_cast<ushort_t*>(x_s);
        __syncthreads();
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf) {
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int wo2 = (lane >> 4) * 4 + r;
                const int ko = wk * 32 + nf * 16 + (lane & 15);
                ystage[(wq * Wo + wo2) * KT + ko] = (short)f2b(acc[nf][r]);
            }
        }
        __syncthreads();
        {
            const int pieces = CP * (KT / 8);
            const size_t base = ((size_t)n * Ho + r0) * Wo;
            for (int i = t; i < pieces; i += 512) {
                const int p = i / (KT / 8);
                const int kg = (i % (KT / 8)) * 8;
                *reinterpret_cast<u32x4*>(
                    y + (base + p) * K + kt * KT + kg) =
                    *reinterpret_cast<const u32x4*>(
                        &ystage[p * KT + kg]);
            }
        }
    }
}

extern "C" int conv3x3_s2_fwd_supported(int H, int W, int C, int K) {
    if (W != 32) return 0;           // Wo == 16
    const int Ho = H / 2;
    if (H % 2 || Ho % S2F_P2) return 0;
    if (C % 32 || C > 128 || K % 64) return 0;
    const int pieces = (2 * S2F_P2 + 1) * W * (C / 8);
    if (pieces > 2560) return 0;     // s2f_issue register budget
    size_t need = ((size_t)64 * 9 * (C + 8)
                   + (size_t)(2 * S2F_P2 + 1) * (2 * (W / 2) + 2)
                     * (C + 8)) * 2;
    return need <= 160 * 1024;
}

extern "C" void launch_conv3x3_s2_fwd(
        const ushort_t* x, const ushort_t* w, ushort_t* y,
        int N, int H, int W, int C, int K, hipStream_t s) {
    const int KT = 64;
    const int LS2 = 2 * (W / 2) + 2;
    const size_t lds_bytes = ((size_t)KT * 9 * (C + 8)
                              + (size_t)(2 * S2F_P2 + 1) * LS2
                                * (C + 8)) * 2;
    static int attr_set = 0;
    if (!attr_set) {
        hipFuncSetAttribute(
            reinterpret_cast<const void*>(&k_conv3x3_s2_fwd),
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = 1;
    }
    const long chunks = (long)N * ((H / 2) / S2F_P2);
    const int nk = K / KT;
    long spread = 256 / nk;
    if (spread > chunks) spread = chunks;
    if (spread < 1) spread = 1;
    hipLaunchKernelGGL(k_conv3x3_s2_fwd, dim3((unsigned)(nk * spread)),
                       dim3(512), lds_bytes, s, x, w, y, N, H, W, C, K,
                       (int)spread);
}

extern "C" int conv3x3_s2_bwd_supported(int Ho, int Wo, int K, int C) {
    if (Ho % S2_P2) return 0;
    if (Wo == 16) {
        // K must hit an instantiated contraction template exactly.
        if ((K != 64 && K != 128 && K != 256) || C % S2_CT) return 0;
        if ((S2_P2 + 1) * Wo * (K / 8) > 1024) return 0;  // reg budget
        size_t need = ((size_t)S2_CT * 9 * (K + 8)
                       + (size_t)(S2_P2 + 1) * (Wo + 8) * (K + 16)) * 2;
        return need <= 160 * 1024;
    }
    if (Wo == 8) {
        if ((K != 128 && K != 256) || C % S2W8_CT) return 0;
        if ((S2_P2 + 1) * Wo * (K / 8) > 1024) return 0;
        size_t need = ((size_t)S2W8_CT * 9 * (K + 8)
                       + (size_t)(S2_P2 + 1) * (Wo + 8) * (K + 16)) * 2;
        return need <= 160 * 1024;
    }
    return 0;
}

extern "C" void launch_conv3x3_s2_bwd(
        const ushort_t* dy, const ushort_t* wt, ushort_t* dx,
        int N, int Ho, int Wo, int K, int C, hipStream_t s) {
    const int CT = (Wo == 8) ? S2W8_CT : S2_CT;
    const int KS = K + 16, KS2 = K + 8, LW = Wo + 8;
    const size_t wbytes = (size_t)CT * 9 * KS2 * 2;
    const size_t dybytes = (size_t)(S2_P2 + 1) * LW * KS * 2;
    int dbuf = 1;
    size_t lds_bytes = wbytes + 2 * dybytes;
    if (lds_bytes > 160 * 1024) {
        dbuf = 0;
        lds_bytes = wbytes + dybytes;
    }
    static int attr_set = 0;
    if (!attr_set) {
        hipFuncSetAttribute(
            reinterpret_cast<const void*>(&k_conv3x3_s2_bwd),
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        hipFuncSetAttribute(
            reinterpret_cast<const void*>(&k_conv3x3_s2_bwd_w8),
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = 1;
    }
    const long chunks = (long)N * (Ho / S2_P2);
    const int nct = C / CT;
    long spread = 256 / nct;
    if (spread > chunks) spread = chunks;
    if (spread < 1) spread = 1;
    if (Wo == 8) {
        hipLaunchKernelGGL(k_conv3x3_s2_bwd_w8,
                           dim3((unsigned)(nct * spread)), dim3(512),
                           lds_bytes, s, dy, wt, dx, N, Ho, K, C,
                           (int)spread, dbuf);
    } else {
        hipLaunchKernelGGL(k_conv3x3_s2_bwd,
                           dim3((unsigned)(nct * spread)), dim3(512),
                           lds_bytes, s, dy, wt, dx, N, Ho, Wo, K, C,
                           (int)spread, dbuf);
    }
}

// =====================================================================
// MFMA 3x3 STRIDE-2 weight gradient (NHWC bf16), experimental.
//
//   dW[k][dh][dw][c] = sum_{n,ho,wo} dy[n,ho,wo,k]
//                      * x[n, 2*ho+dh-1, 2*wo+dw-1, c]
//
// Same implicit-GEMM structure as the stride-1 wrw kernel (transposed
// dy_t/x_t staging, 36 accumulator fragments per wave, 5/4 tau-split
// wave pairs, split-K fp32 slabs + the SAME k_wrw_reduce), with the
// x side staged POLYPHASE: each of the 2P+1 input lines of a chunk is
// stored as [E: even cols (Wo)][4 zero pads][O: odd cols (Wo)][4 pads]
// (row stride LS2 = 2*Wo+8), so the dw=1 tap is one aligned b128 read
// (E) and dw=0/2 come from one aligned 16-value O window via the
// stride-1 kernel's byte-rotation extracts (dw=0 -> shift 6, dw=2 ->
// shift 8).  x global loads pair SAME-PARITY pixels (w, w+2) so the
// packed b32 transpose writes stay adjacent in E/O.  The x_t stride is
// runtime (content (2P+1)*LS2 > the stride-1 search's 288): the
// launcher picks a multiple of 8 whose dword stride has gcd 4 with the
// 64 LDS banks (16 distinct banks per 16-lane fragment group); no XOR
// swizzle in v1 (measure first).  dy_t reuses DY_STRIDE + DSWZ as-is.
// Constraints: dy Wo in {8, 16}, Ho % 4 == 0, C % 64, K % 64.
// Experimental: gated off until an in-context A/B vs MIOpen's s2 wrw.
// =====================================================================

#define S2W_P 4

struct S2WRegs {
    V16 vdy[1][2];
    V16 vx[3][2];
};

__device__ __forceinline__ void s2w_issue(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
        int n, int lo0, int Ho, int Wo, int C, int K, int kt, int ct,
        int CP, int t, S2WRegs& r) {
    const int dyn8 = CP * 4;         // pixel-pairs x 8 k-groups
    #pragma unroll
    for (int it = 0; it < 1; ++it) {
        const int i = t + it * 512;
        if (i < dyn8) {
            const int p = (i >> 3) * 2;
            const ushort_t* g = dy +
                (((size_t)n * Ho + (lo0 + p / Wo)) * Wo + p % Wo) * K
                + (size_t)kt * 64 + (i & 7) * 8;
            r.vdy[it][0].u4 = *reinterpret_cast<const u32x4*>(g);
            r.vdy[it][1].u4 = *reinterpret_cast<const u32x4*>(g + K);
        }
    }
    const int Hi = 2 * Ho, Wi = 2 * Wo;
    const int xn8 = (2 * S2W_P + 1) * (Wi / 2) * 8;
    #pragma unroll
    for (int it = 0; it < 3; ++it) {
        const int i = t + it * 512;
        if (i < xn8) {
            const int j = i / ((Wi / 2) * 8);
            const int rem = i % ((Wi / 2) * 8);
            const int pr = rem >> 3;
            const int wbase = (pr & 1) + 4 * (pr >> 1);
            const int h = 2 * lo0 - 1 + j;
            if (h < 0 || h >= Hi) {
                r.vx[it][0].u4 = u32x4{0u, 0u, 0u, 0u};
                r.vx[it][1].u4 = u32x4{0u, 0u, 0u, 0u};
            } else {
                const ushort_t* g = x +
                    (((size_t)n * Hi + h) * Wi + wbase) * C
                    + (size_t)ct * 64 + (rem & 7) * 8;
                r.vx[it][0].u4 = *reinterpret_cast<const u32x4*>(g);
                r.vx[it][1].u4 =
                    *reinterpret_cast<const u32x4*>(g + 2 * C);
            }
        }
    }
}

__device__ __forceinline__ void s2w_write(
        short* __restrict__ dy_t, short* __restrict__ x_t,
        int Wo, int CP, int XT2S, int LS2, int t, S2WRegs& r) {
    const int dyn8 = CP * 4;
    #pragma unroll
    for (int it = 0; it < 1; ++it) {
        const int i = t + it * 512;
        if (i < dyn8) {
            const int p = (i >> 3) * 2;
            const int kg = (i & 7) * 8;
            #pragma unroll
            for (int j = 0; j < 8; ++j) {
                const unsigned int packed =
                    (unsigned int)(unsigned short)r.vdy[it][0].s[j]
                    | ((unsigned int)(unsigned short)
                       r.vdy[it][1].s[j] << 16);
                *reinterpret_cast<unsigned int*>(
                    &dy_t[(kg + j) * DY_STRIDE
                          + DSWZ(kg + j, p)]) = packed;
            }
        }
    }
    const int Wi = 2 * Wo;
    const int xn8 = (2 * S2W_P + 1) * (Wi / 2) * 8;
    #pragma unroll
    for (int it = 0; it < 3; ++it) {
        const int i = t + it * 512;
        if (i < xn8) {
            const int j = i / ((Wi / 2) * 8);
            const int rem = i % ((Wi / 2) * 8);
            const int pr = rem >> 3;
            const int cg = (rem & 7) * 8;
            const int wbase = (pr & 1) + 4 * (pr >> 1);
            // same-parity pair (wbase, wbase+2) -> adjacent E/O slots
            const int col = (wbase & 1)
                ? (Wo + 4 + ((wbase - 1) >> 1))   // O section
                : (wbase >> 1);                   // E section
            #pragma unroll
            for (int jj = 0; jj < 8; ++jj) {
                const unsigned int packed =
                    (unsigned int)(unsigned short)r.vx[it][0].s[jj]
                    | ((unsigned int)(unsigned short)
                       r.vx[it][1].s[jj] << 16);
                *reinterpret_cast<unsigned int*>(
                    &x_t[(cg + jj) * XT2S + j * LS2 + col]) = packed;
            }
        }
    }
}

template <int WT2>
__device__ __forceinline__ void s2w_mfma_phase(
        const short* __restrict__ dy_t, const short* __restrict__ x_t,
        int kchunks, int Wo, int XT2S, int LS2, int wk, int wc,
        int row16, int slot8, f32x4 (&acc)[2][2][5]) {
    for (int kc = 0; kc < kchunks; ++kc) {
        const int p0 = kc * 32 + slot8;
        const int li = p0 / Wo;
        const int w0 = p0 % Wo;
        bf16x8 afrag[2];
        #pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            const int row = wk * 32 + mf * 16 + row16;
            afrag[mf] = *reinterpret_cast<const bf16x8*>(
                &dy_t[row * DY_STRIDE + DSWZ(row, p0)]);
        }
        int ti = 0;
        #pragma unroll
        for (int g = 0; g < 2; ++g) {
            constexpr int dh0 = WT2 ? 1 : 0;
            constexpr int dh1 = WT2 ? 2 : 1;
            const int dh = g ? dh1 : dh0;
            constexpr int lo0 = WT2 ? 2 : 0;
            constexpr int hi1 = WT2 ? 3 : 2;
            const int lo = g ? 0 : lo0;
            const int hi = g ? hi1 : 3;
            const int slot = 2 * li + dh;    // input line slot (halo 0)
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf) {
                const int crow = wc * 32 + nf * 16 + row16;
                const short* xrow = &x_t[crow * XT2S + slot * LS2];
                // E: the dw=1 tap, one aligned b128
                bf16x8 efrag;
                if (lo <= 1 && 1 < hi)
                    efrag = *reinterpret_cast<const bf16x8*>(
                        &xrow[w0]);
                // O window: taps dw=0 (shift 6) and dw=2 (shift 8)
                V16 vlo, vhi;
                if (lo < 1 || hi > 2) {
                    vlo.u4 = *reinterpret_cast<const u32x4*>(
                        &xrow[Wo + w0]);
                    vhi.u4 = *reinterpret_cast<const u32x4*>(
                        &xrow[Wo + w0 + 8]);
                }
                unsigned int win[8] = {vlo.u[0], vlo.u[1], vlo.u[2],
                                       vlo.u[3], vhi.u[0], vhi.u[1],
                                       vhi.u[2], vhi.u[3]};
                int tj = ti;
                #pragma unroll
                for (int dw = 0; dw < 3; ++dw) {
                    if (dw < lo || dw >= hi) continue;
                    bf16x8 bfrag;
                    if (dw == 1) {
                        bfrag = efrag;
                    } else {
                        u32x4 frag;
                        const int sh = (dw == 0) ? 6 : 8;
                        const int d0 = sh >> 2;
                        const int rem2 = sh & 3;
                        if (rem2 == 0) {
                            frag[0] = win[d0]; frag[1] = win[d0 + 1];
                            frag[2] = win[d0 + 2];
                            frag[3] = win[d0 + 3];
                        } else {
                            frag[0] = __builtin_amdgcn_alignbyte(
                                win[d0 + 1], win[d0], rem2);
                            frag[1] = __builtin_amdgcn_alignbyte(
                                win[d0 + 2], win[d0 + 1], rem2);
                            frag[2] = __builtin_amdgcn_alignbyte(
                                win[d0 + 3], win[d0 + 2], rem2);
                            frag[3] = __builtin_amdgcn_alignbyte(
                                win[d0 + 4], win[d0 + 3], rem2);
                        }
                        bfrag = *reinterpret_cast<const bf16x8*>(&frag);
                    }
                    #pragma unroll
                    for (int mf = 0; mf < 2; ++mf)
                        acc[mf][nf][tj] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                afrag[mf], bfrag, acc[mf][nf][tj],
                                0, 0, 0);
                    ++tj;
                }
            }
            ti += (g ? hi1 : 3) - (g ? 0 : lo0);
        }
    }
}

extern "C" __global__ __launch_bounds__(512, 2) void k_conv3x3_s2_wrw(
        const ushort_t* __restrict__ x, const ushort_t* __restrict__ dy,
        float* __restrict__ ws, int N, int Ho, int Wo, int C, int K,
        int XT2S, int nsplit) {
    extern __shared__ short lds[];
    const int LS2 = 2 * Wo + 8;
    const int CP = S2W_P * Wo;
    const int kchunks = CP / 32;
    const int BUF = 64 * DY_STRIDE + 64 * XT2S;
    const int nc = C / 64;
    const int tile = blockIdx.x / nsplit;
    const int split = blockIdx.x % nsplit;
    const int kt = tile / nc;
    const int ct = tile % nc;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;
    const int wk = wid >> 2;
    const int wc = (wid >> 1) & 1;
    const int wt2 = wid & 1;
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;

    f32x4 acc[2][2][5];
    #pragma unroll
    for (int mf = 0; mf < 2; ++mf)
        #pragma unroll
        for (int nf = 0; nf < 2; ++nf)
            #pragma unroll
            for (int ti = 0; ti < 5; ++ti)
                #pragma unroll
                for (int r = 0; r < 4; ++r)
                    acc[mf][nf][ti][r] = 0.f;

    const int rows_per_img = Ho / S2W_P;
    const long chunks_total = (long)N * rows_per_img;

    // Zero the O-section front pads (cols Wo..Wo+3: the j=-1 slot read
    // by the dw=0 tap at w0=0) of both x_t buffers once.
    for (int b = 0; b < 2; ++b) {
        short* x_t = lds + b * BUF + 64 * DY_STRIDE;
        for (int i = t; i < 64 * (2 * S2W_P + 1) * 2; i += 512) {
            const int cc = i / ((2 * S2W_P + 1) * 2);
            const int rem = i % ((2 * S2W_P + 1) * 2);
            const int j = rem >> 1;
            const int half = (rem & 1) * 2;
            *reinterpret_cast<unsigned int*>(
                &x_t[cc * XT2S + j * LS2 + Wo + half]) = 0u;
        }
    }

    S2WRegs regs;
    #define S2W_NH(qq) \
        const int n_ = (int)((qq) / rows_per_img); \
        const int lo_ = (int)((qq) % rows_per_img) * S2W_P;

    if (split < chunks_total) {
        S2W_NH(split)
        s2w_issue(x, dy, n_, lo_, Ho, Wo, C, K, kt, ct, CP, t, regs);
        s2w_write(lds, lds + 64 * DY_STRIDE, Wo, CP, XT2S, LS2, t, regs);
    }
    __syncthreads();
    if (split + nsplit < chunks_total) {
        S2W_NH(split + nsplit)
        s2w_issue(x, dy, n_, lo_, Ho, Wo, C, K, kt, ct, CP, t, regs);
    }

    int cur = 0;
    for (long q = split; q < chunks_total; q += nsplit, cur ^= 1) {
        const short* dy_t = lds + cur * BUF;
        const short* x_t = dy_t + 64 * DY_STRIDE;

        if (wt2 == 0)
            s2w_mfma_phase<0>(dy_t, x_t, kchunks, Wo, XT2S, LS2, wk, wc,
                              row16, slot8, acc);
        else
            s2w_mfma_phase<1>(dy_t, x_t, kchunks, Wo, XT2S, LS2, wk, wc,
                              row16, slot8, acc);

        if (q + nsplit < chunks_total) {
            short* ndy = lds + (cur ^ 1) * BUF;
            s2w_write(ndy, ndy + 64 * DY_STRIDE, Wo, CP, XT2S, LS2, t,
                      regs);
            if (q + 2 * nsplit < chunks_total) {
                S2W_NH(q + 2 * nsplit)
                s2w_issue(x, dy, n_, lo_, Ho, Wo, C, K, kt, ct, CP, t,
                          regs);
            }
        }
        __syncthreads();
    }
    #undef S2W_NH

    // fp32 partial slab (same layout as the stride-1 wrw epilogue)
    float* slab = ws + (size_t)split * K * 9 * C;
    #pragma unroll
    for (int ti = 0; ti < 5; ++ti) {
        if (wt2 && ti >= 4)
            continue;
        const int tau = (wt2 ? 5 : 0) + ti;
        #pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
            #pragma unroll
            for (int nf = 0; nf < 2; ++nf) {
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int klocal = wk * 32 + mf * 16
                        + ((lane >> 4) * 4 + r);
                    const int clocal = wc * 32 + nf * 16
                        + (lane & 15);
                    slab[((size_t)(kt * 64 + klocal) * 9 + tau) * C
                         + ct * 64 + clocal] = acc[mf][nf][ti][r];
                }
            }
        }
    }
}

static int s2w_xt_stride(int Wo) {
    // content (2P+1)*LS2; stride mult of 8 whose dword count has
    // gcd(., 64) == 4 -> 16 distinct banks across a 16-lane fragment
    const int content = (2 * S2W_P + 1) * (2 * Wo + 8);
    int s = (content + 7) / 8 * 8;
    while ((s / 2) % 4 != 0 || ((s / 2) % 64) % 8 != 4) {
        s += 8;
        if (s > content + 128) break;   // give up: take mult-of-8
    }
    return s;
}

extern "C" int conv3x3_s2_wrw_supported(int Ho, int Wo, int C, int K) {
    if (Wo != 8 && Wo != 16) return 0;
    if (Ho % S2W_P) return 0;
    if (C % 64 || K % 64) return 0;
    const int XT2S = s2w_xt_stride(Wo);
    size_t lds_bytes = 2 * (size_t)(64 * DY_STRIDE + 64 * XT2S) * 2;
    if (lds_bytes > 160 * 1024) return 0;
    const int xn8 = (2 * S2W_P + 1) * Wo * 8;  // Wi/2 == Wo
    if (xn8 > 1536) return 0;        // s2w_issue register budget
    return 1;
}

extern "C" int conv3x3_s2_wrw_nsplit(int N, int Ho, int Wo, int C,
                                     int K) {
    long chunks = (long)N * (Ho / S2W_P);
    int tiles = (K / 64) * (C / 64);
    long target = 256 / tiles;
    if (target < 1) target = 1;
    if (target > chunks) target = chunks;
    return (int)target;
}

extern "C" void launch_conv3x3_s2_wrw(
        const ushort_t* x, const ushort_t* dy, float* ws, float* dw,
        int N, int Ho, int Wo, int C, int K, hipStream_t s) {
    const int XT2S = s2w_xt_stride(Wo);
    const int nsplit = conv3x3_s2_wrw_nsplit(N, Ho, Wo, C, K);
    const int tiles = (K / 64) * (C / 64);
    const size_t lds_bytes = 2 * (size_t)(64 * DY_STRIDE + 64 * XT2S)
        * 2;
    static int attr_set = 0;
    if (!attr_set) {
        hipFuncSetAttribute(
            reinterpret_cast<const void*>(&k_conv3x3_s2_wrw),
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = 1;
    }
    hipLaunchKernelGGL(k_conv3x3_s2_wrw, dim3(tiles * nsplit), dim3(512),
                       lds_bytes, s, x, dy, ws, N, Ho, Wo, C, K, XT2S,
                       nsplit);
    const long n = (long)K * 9 * C;
    const int ngroups = (nsplit + 15) / 16;
    const long nchunks = (n + 1023) / 1024;
    if (ngroups > 1)
        hipMemsetAsync(dw, 0, n * sizeof(float), s);
    long blocks = nchunks * ngroups;
    if (blocks > 4096) blocks = 4096;
    hipLaunchKernelGGL(k_wrw_reduce, dim3((unsigned)blocks), dim3(256), 0,
                       s, ws, dw, n, nsplit, ngroups, nchunks);
}

// =====================================================================
// Wo == 8 stride-2 backward-data, SECOND attempt (experimental).
//
// The shipped k_conv3x3_s2_bwd_w8 measured a 3.8 ms/step regression in
// context (profiles/bench_r10): P2=2 makes 4096 tiny 64-pixel chunks
// with 5 barriers + an f32 merge each.  This variant uses P2=4
// (128-pixel chunks, half the barrier rounds, 2x work per round) with
// frag-quad wave groups ({ph00, ph11} x both row-pairs vs the two
// mixed phases: 10 vs 8 MFMAs per cc, 1.25x balance) and keeps the
// 4-way k-split + f32 LDS merge.  Single-buffered dy window with
// register prefetch (76 KB weights + 43.5 KB window; two windows do
// not fit).  Gated: numerics test requires ADAPTDL_EXPERIMENTAL_S2_FWD,
// production use requires winning a round-2 in-context A/B.
// =====================================================================

#define S2W8B_P2 4

struct S2W8bRegs {
    V16 v[3];
};

__device__ __forceinline__ void s2w8b_issue(
        const ushort_t* __restrict__ dy, int n, int r0, int Ho, int Wo,
        int K, int t, S2W8bRegs& r) {
    const int pieces = (S2W8B_P2 + 1) * Wo * (K / 8);
    #pragma unroll
    for (int it = 0; it < 3; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (Wo * (K / 8));
            const int rem = i % (Wo * (K / 8));
            const int wo = rem / (K / 8);
            const int kg = (rem % (K / 8)) * 8;
            const int ho = r0 + j;
            if (ho >= Ho) {
                r.v[it].u4 = u32x4{0u, 0u, 0u, 0u};
            } else {
                r.v[it].u4 = *reinterpret_cast<const u32x4*>(
                    dy + (((size_t)n * Ho + ho) * Wo + wo) * K + kg);
            }
        }
    }
}

__device__ __forceinline__ void s2w8b_write(
        short* __restrict__ dy_s, int Wo, int K, int KS, int LW, int t,
        S2W8bRegs& r) {
    const int pieces = (S2W8B_P2 + 1) * Wo * (K / 8);
    #pragma unroll
    for (int it = 0; it < 3; ++it) {
        const int i = t + it * 512;
        if (i < pieces) {
            const int j = i / (Wo * (K / 8));
            const int rem = i % (Wo * (K / 8));
            const int wo = rem / (K / 8);
            const int kg = (rem % (K / 8)) * 8;
            *reinterpret_cast<u32x4*>(
                &dy_s[(j * LW + wo) * KS + kg]) = r.v[it].u4;
        }
    }
    const int zp = (S2W8B_P2 + 1) * 8 * (K / 8);   // pad col refresh
    for (int i = t; i < zp; i += 512) {
        const int j = i / (8 * (K / 8));
        const int rem = i % (8 * (K / 8));
        const int pc = rem / (K / 8);
        const int kg = (rem % (K / 8)) * 8;
        u32x4 z = {0u, 0u, 0u, 0u};
        *reinterpret_cast<u32x4*>(
            &dy_s[(j * LW + Wo + pc) * KS + kg]) = z;
    }
}

// Wave group G owns four fragments: G=0 -> phases {0,3} x row-pairs,
// G=1 -> {1,2} x row-pairs.  KQ = k-quarter, KC = per-wave cc count.
template <int G, int KQ, int KC>
__device__ __forceinline__ void s2w8b_contract(
        const short* __restrict__ dy_s, const short* __restrict__ w_lds,
        int KS, int KS2, int LW, int row16, int slot8,
        f32x4 (&acc)[4][2]) {
    const int w2 = row16 & 7;
    const int h2base = row16 >> 3;   // 0..1 within the row pair
    #pragma unroll
    for (int cc = 0; cc < KC; ++cc) {
        const int kbase = KQ * (KC * 32) + cc * 32;
        #pragma unroll
        for (int fi = 0; fi < 4; ++fi) {
            constexpr int PHA = (G == 0) ? 0 : 1;
            constexpr int PHB = (G == 0) ? 3 : 2;
            const int ph = (fi >> 1) ? PHB : PHA;
            const int f = fi & 1;                // row pair
            const int h2 = 2 * f + h2base;
            const int a = ph >> 1, b = ph & 1;
            #pragma unroll
            for (int dh = 0; dh < 3; ++dh) {
                if (a ? (dh == 1) : (dh != 1)) continue;
                #pragma unroll
                for (int dw = 0; dw < 3; ++dw) {
                    if (b ? (dw == 1) : (dw != 1)) continue;
                    const int j = h2 + (dh == 0 ? 1 : 0);
                    const int col = w2 + (dw == 0 ? 1 : 0);
                    const bf16x8 afrag =
                        *reinterpret_cast<const bf16x8*>(
                            &dy_s[(j * LW + col) * KS + kbase + slot8]);
                    const int tau = dh * 3 + dw;
                    const bf16x8 bfrag =
                        *reinterpret_cast<const bf16x8*>(
                            &w_lds[(row16 * 9 + tau) * KS2 + kbase
                                   + slot8]);
                    acc[fi][cc & 1] =
                        __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            afrag, bfrag, acc[fi][cc & 1], 0, 0, 0);
                }
            }
        }
    }
}

template <int KC>
__device__ __forceinline__ void s2w8b_contract_g(
        int wid7, const short* __restrict__ dy_s,
        const short* __restrict__ w_lds, int KS, int KS2, int LW,
        int row16, int slot8, f32x4 (&acc)[4][2]) {
    switch (wid7) {
    case 0: s2w8b_contract<0, 0, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    case 1: s2w8b_contract<0, 1, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    case 2: s2w8b_contract<0, 2, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    case 3: s2w8b_contract<0, 3, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    case 4: s2w8b_contract<1, 0, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    case 5: s2w8b_contract<1, 1, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    case 6: s2w8b_contract<1, 2, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                     slot8, acc); break;
    default: s2w8b_contract<1, 3, KC>(dy_s, w_lds, KS, KS2, LW, row16,
                                      slot8, acc); break;
    }
}

extern "C" __global__ __launch_bounds__(512, 2)
void k_conv3x3_s2_bwd_w8b(
        const ushort_t* __restrict__ dy, const ushort_t* __restrict__ wt,
        ushort_t* __restrict__ dx, int N, int Ho, int K, int C,
        int spread) {
    extern __shared__ short lds[];
    const int Wo = 8;
    const int KS = K + 16;
    const int KS2 = K + 8;
    const int LW = Wo + 8;
    short* w_lds = lds;              // [S2W8_CT * 9][KS2]
    short* dy_s = lds + S2W8_CT * 9 * KS2;   // single buffer

    const int Hi = 2 * Ho, Wi = 2 * Wo;
    const int ct = blockIdx.x / spread;
    const int sp = blockIdx.x % spread;

    const int t = threadIdx.x;
    const int lane = t & 63;
    const int wid = t >> 6;
    const int wg2 = wid >> 2;        // frag-quad group
    const int row16 = lane & 15;
    const int slot8 = (lane >> 4) * 8;
    const int CP2 = 4 * S2W8B_P2 * Wo;   // 128 chunk pixels

    {
        const int pieces = S2W8_CT * 9 * (K / 8);
        const ushort_t* wg = wt + (size_t)ct * S2W8_CT * 9 * K;
        for (int i = t; i < pieces; i += 512) {
            const int row = i / (K / 8);
            const int kg = (i % (K / 8)) * 8;
            *reinterpret_cast<u32x4*>(&w_lds[row * KS2 + kg]) =
                *reinterpret_cast<const u32x4*>(
                    wg + (size_t)row * K + kg);
        }
    }

    const int rows_per_img = Ho / S2W8B_P2;
    const long chunks_total = (long)N * rows_per_img;

    S2W8bRegs regs;
    if (sp < chunks_total) {
        s2w8b_issue(dy, (int)(sp / rows_per_img),
                    (int)(sp % rows_per_img) * S2W8B_P2, Ho, Wo, K, t,
                    regs);
        s2w8b_write(dy_s, Wo, K, KS, LW, t, regs);
    }
    __syncthreads();
    if (sp + spread < chunks_total) {
        const long qn = sp + spread;
        s2w8b_issue(dy, (int)(qn / rows_per_img),
                    (int)(qn % rows_per_img) * S2W8B_P2, Ho, Wo, K, t,
                    regs);
    }

    for (long q = sp; q < chunks_total; q += spread) {
        const int n = (int)(q / rows_per_img);
        const int r0 = (int)(q % rows_per_img) * S2W8B_P2;

        if (q != sp) {
            __syncthreads();         // merge/store of q-1 fully read
            s2w8b_write(dy_s, Wo, K, KS, LW, t, regs);
            __syncthreads();
            if (q + spread < chunks_total) {
                const long qn = q + spread;
                s2w8b_issue(dy, (int)(qn / rows_per_img),
                            (int)(qn % rows_per_img) * S2W8B_P2, Ho, Wo,
                            K, t, regs);
            }
        }

        f32x4 acc[4][2];
        #pragma unroll
        for (int fi = 0; fi < 4; ++fi)
            #pragma unroll
            for (int pp = 0; pp < 2; ++pp)
                #pragma unroll
                for (int r = 0; r < 4; ++r)
                    acc[fi][pp][r] = 0.f;

        if (K / 32 == 8)
            s2w8b_contract_g<2>(wid & 7, dy_s, w_lds, KS, KS2, LW,
                                row16, slot8, acc);
        else
            s2w8b_contract_g<1>(wid & 7, dy_s, w_lds, KS, KS2, LW,
                                row16, slot8, acc);

        float* fstage = reinterpret_cast<float*>(dy_s);  // [128][CT]
        __syncthreads();
        for (int i = t; i < CP2 * S2W8_CT; i += 512)
            fstage[i] = 0.f;
        __syncthreads();
        #pragma unroll
        for (int fi = 0; fi < 4; ++fi) {
            const int ph = (fi >> 1) ? ((wg2 == 0) ? 3 : 2)
                                     : ((wg2 == 0) ? 0 : 1);
            const int f = fi & 1;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = (lane >> 4) * 4 + r;
                const int h2 = 2 * f + (m >> 3);
                const int w2 = m & 7;
                const int c = lane & 15;
                const int s = (ph * S2W8B_P2 + h2) * Wo + w2;
                atomicAdd(&fstage[s * S2W8_CT + c],
                          acc[fi][0][r] + acc[fi][1][r]);
            }
        }
        __syncthreads();
        {
            const int pieces = 2 * S2W8B_P2 * Wi * (S2W8_CT / 8);
            const size_t base = ((size_t)n * Hi + 2 * r0) * Wi;
            for (int i = t; i < pieces; i += 512) {
                const int hl = i / (Wi * (S2W8_CT / 8));
                const int rem = i % (Wi * (S2W8_CT / 8));
                const int wi = rem / (S2W8_CT / 8);
                const int cg = (rem % (S2W8_CT / 8)) * 8;
                const int ph = ((hl & 1) << 1) | (wi & 1);
                const int s = (ph * S2W8B_P2 + (hl >> 1)) * Wo
                    + (wi >> 1);
                ushort_t out[8];
                #pragma unroll
                for (int jj = 0; jj < 8; ++jj)
                    out[jj] = f2b(fstage[s * S2W8_CT + cg + jj]);
                *reinterpret_cast<u32x4*>(
                    dx + (base + (size_t)hl * Wi + wi) * C
                    + ct * S2W8_CT + cg) =
                    *reinterpret_cast<const u32x4*>(out);
            }
        }
    }
}

extern "C" int conv3x3_s2_bwd_w8b_supported(int Ho, int Wo, int K,
                                            int C) {
    if (Wo != 8) return 0;
    if (Ho % S2W8B_P2) return 0;
    if ((K != 128 && K != 256) || C % S2W8_CT) return 0;
    if ((S2W8B_P2 + 1) * Wo * (K / 8) > 1536) return 0;
    size_t need = ((size_t)S2W8_CT * 9 * (K + 8)
                   + (size_t)(S2W8B_P2 + 1) * (Wo + 8) * (K + 16)) * 2;
    return need <= 160 * 1024;
}

extern "C" void launch_conv3x3_s2_bwd_w8b(
        const ushort_t* dy, const ushort_t* wt, ushort_t* dx,
        int N, int Ho, int K, int C, hipStream_t s) {
    const int Wo = 8;
    const size_t lds_bytes = ((size_t)S2W8_CT * 9 * (K + 8)
                              + (size_t)(S2W8B_P2 + 1) * (Wo + 8)
                                * (K + 16)) * 2;
    static int attr_set = 0;
    if (!attr_set) {
        hipFuncSetAttribute(
            reinterpret_cast<const void*>(&k_conv3x3_s2_bwd_w8b),
            hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
        attr_set = 1;
    }
    const long chunks = (long)N * (Ho / S2W8B_P2);
    const int nct = C / S2W8_CT;
    long spread = 256 / nct;
    if (spread > chunks) spread = chunks;
    if (spread < 1) spread = 1;
    hipLaunchKernelGGL(k_conv3x3_s2_bwd_w8b,
                       dim3((unsigned)(nct * spread)), dim3(512),
                       lds_bytes, s, dy, wt, dx, N, Ho, K, C,
                       (int)spread);
}
