// Empirically derive the A/B/D lane<->element mappings of
// v_mfma_f32_16x16x32_bf16 and v_mfma_f32_32x32x16_bf16 on gfx950.
//
// For each basis position (L,E) of one operand, the other operand is
// value-encoded TWICE (value = lane+1, then value = elem+1 — both exact
// in bf16), so every nonzero D entry identifies the exact encoded
// position it came from.  tools/mfma_decode.py reconstructs
// Amap/Bmap/Dmap and emits index formulas.
//
// Build: hipcc --offload-arch=gfx950 -O2 tools/mfma_probe.hip -o probe

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ __forceinline__ short f2b(float f) {
    __hip_bfloat16 h = __float2bfloat16(f);
    return *reinterpret_cast<short*>(&h);
}

// enc==0: value = lane+1; enc==1: value = elem+1 (exact in bf16).
template <int REGS, typename DVEC, typename MFMA>
__device__ void run_probes(float* outA, float* outB, MFMA mfma) {
    const int lane = threadIdx.x;
    for (int enc = 0; enc < 2; ++enc) {
        for (int L = 0; L < 64; ++L) {
            for (int E = 0; E < 8; ++E) {
                bf16x8 a = {0, 0, 0, 0, 0, 0, 0, 0};
                if (lane == L) a[E] = f2b(1.0f);
                bf16x8 b;
                for (int e = 0; e < 8; ++e)
                    b[e] = f2b((float)((enc ? e : lane) + 1));
                DVEC d;
                for (int r = 0; r < REGS; ++r) d[r] = 0.f;
                d = mfma(a, b, d);
                const size_t probe = (size_t)enc * 512 + L * 8 + E;
                for (int r = 0; r < REGS; ++r)
                    outA[(probe * 64 + lane) * REGS + r] = d[r];
            }
        }
        for (int L = 0; L < 64; ++L) {
            for (int E = 0; E < 8; ++E) {
                bf16x8 b = {0, 0, 0, 0, 0, 0, 0, 0};
                if (lane == L) b[E] = f2b(1.0f);
                bf16x8 a;
                for (int e = 0; e < 8; ++e)
                    a[e] = f2b((float)((enc ? e : lane) + 1));
                DVEC d;
                for (int r = 0; r < REGS; ++r) d[r] = 0.f;
                d = mfma(a, b, d);
                const size_t probe = (size_t)enc * 512 + L * 8 + E;
                for (int r = 0; r < REGS; ++r)
                    outB[(probe * 64 + lane) * REGS + r] = d[r];
            }
        }
    }
}

struct M16 {
    __device__ f32x4 operator()(bf16x8 a, bf16x8 b, f32x4 d) const {
        return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, d, 0, 0, 0);
    }
};
struct M32 {
    __device__ f32x16 operator()(bf16x8 a, bf16x8 b, f32x16 d) const {
        return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, d, 0, 0, 0);
    }
};

extern "C" __global__ void probe16(float* oa, float* ob) {
    run_probes<4, f32x4>(oa, ob, M16());
}
extern "C" __global__ void probe32(float* oa, float* ob) {
    run_probes<16, f32x16>(oa, ob, M32());
}

static void dump(const char* name, const float* A, const float* B,
                 int regs) {
    printf("==== %s ====\n", name);
    const float* bufs[2] = {A, B};
    const char* tag = "AB";
    for (int t = 0; t < 2; ++t)
        for (int enc = 0; enc < 2; ++enc)
            for (int p = 0; p < 512; ++p)
                for (int l = 0; l < 64; ++l)
                    for (int r = 0; r < regs; ++r) {
                        size_t i = (((size_t)enc * 512 + p) * 64 + l)
                            * regs + r;
                        if (bufs[t][i] != 0.0f)
                            printf("%c %d %d %d %d %d %.0f\n", tag[t],
                                   enc, p / 8, p % 8, l, r, bufs[t][i]);
                    }
}

int main() {
    size_t n16 = 1024L * 64 * 4, n32 = 1024L * 64 * 16;
    float *a16, *b16, *a32, *b32;
    (void)hipMalloc(&a16, n16 * 4);
    (void)hipMalloc(&b16, n16 * 4);
    (void)hipMalloc(&a32, n32 * 4);
    (void)hipMalloc(&b32, n32 * 4);
    (void)hipMemset(a16, 0, n16 * 4);
    (void)hipMemset(b16, 0, n16 * 4);
    (void)hipMemset(a32, 0, n32 * 4);
    (void)hipMemset(b32, 0, n32 * 4);
    hipLaunchKernelGGL(probe16, dim3(1), dim3(64), 0, 0, a16, b16);
    hipLaunchKernelGGL(probe32, dim3(1), dim3(64), 0, 0, a32, b32);
    (void)hipDeviceSynchronize();
    float* h = (float*)malloc(n32 * 4);
    float* h2 = (float*)malloc(n32 * 4);
    (void)hipMemcpy(h, a16, n16 * 4, hipMemcpyDeviceToHost);
    (void)hipMemcpy(h2, b16, n16 * 4, hipMemcpyDeviceToHost);
    dump("mfma_f32_16x16x32_bf16", h, h2, 4);
    (void)hipMemcpy(h, a32, n32 * 4, hipMemcpyDeviceToHost);
    (void)hipMemcpy(h2, b32, n32 * 4, hipMemcpyDeviceToHost);
    dump("mfma_f32_32x32x16_bf16", h, h2, 16);
    printf("PROBE_DONE\n");
    return 0;
}
