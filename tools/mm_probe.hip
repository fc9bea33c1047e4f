// Phase probe for k_conv3x3_mm (build with/without -DMM_PROBE_NOSTORE).
#include "../adaptdl_amd/ops/hip/conv_kernels.hip"
#include <cstdio>

int main() {
    const int N = 1024, H = 32, W = 32, C = 64, K = 64;
    size_t xs = (size_t)N * H * W * C * 2, ws_ = (size_t)K * 9 * C * 2;
    unsigned short *x, *w, *y;
    (void)hipMalloc(&x, xs);
    (void)hipMalloc(&w, ws_);
    (void)hipMalloc(&y, xs);
    (void)hipMemset(x, 0x3c, xs);
    (void)hipMemset(w, 0x3c, ws_);
    hipEvent_t e0, e1;
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    for (int i = 0; i < 3; ++i)
        launch_conv3x3_mm(x, w, y, N, H, W, C, K, 0);
    (void)hipDeviceSynchronize();
    (void)hipEventRecord(e0, 0);
    for (int i = 0; i < 20; ++i)
        launch_conv3x3_mm(x, w, y, N, H, W, C, K, 0);
    (void)hipEventRecord(e1, 0);
    (void)hipDeviceSynchronize();
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    printf("per-call us: %.1f\n", ms * 1000 / 20);
    return 0;
}
