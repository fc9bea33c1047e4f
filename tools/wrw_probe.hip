// Phase-isolation probe for k_conv3x3_wrw: builds the kernel in three
// variants (full / no-loads / no-mfma) and times each on layer1 shapes.
// Build three ways:
//   hipcc --offload-arch=gfx950 -O3 tools/wrw_probe.hip -o p_full
//   hipcc ... -DWRW_PROBE_NOLOAD -o p_noload
//   hipcc ... -DWRW_PROBE_NOMFMA -o p_nomfma
#include "../adaptdl_amd/ops/hip/conv_kernels.hip"
#include <cstdio>

int main() {
    const int N = 1024, H = 32, W = 32, C = 64, K = 64;
    size_t xs = (size_t)N * H * W * C * 2, ds = (size_t)N * H * W * K * 2;
    unsigned short *x, *dy;
    float *ws, *dw;
    (void)hipMalloc(&x, xs);
    (void)hipMalloc(&dy, ds);
    int nsplit = conv3x3_wrw_nsplit(N, H, W, C, K);
    (void)hipMalloc(&ws, (size_t)nsplit * K * 9 * C * 4);
    (void)hipMalloc(&dw, (size_t)K * 9 * C * 4);
    (void)hipMemset(x, 0x3c, xs);
    (void)hipMemset(dy, 0x3c, ds);
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    for (int i = 0; i < 3; ++i)
        launch_conv3x3_wrw(x, dy, ws, dw, N, H, W, C, K, 0);
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0, 0);
    for (int i = 0; i < 20; ++i)
        launch_conv3x3_wrw(x, dy, ws, dw, N, H, W, C, K, 0);
    (void)hipEventRecord(e1, 0);
    (void)hipDeviceSynchronize();
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    printf("per-call us: %.1f\n", ms * 1000 / 20);
    return 0;
}
