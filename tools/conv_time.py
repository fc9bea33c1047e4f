import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch.nn.functional as F
from adaptdl_amd import ops
ext = ops._load_extension()
dev = torch.device("cuda")
shapes = [(1024, 64, 32, 32, 64), (1024, 128, 16, 16, 128)]
for n, c, h, w, k in shapes:
    x = torch.randn(n, c, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    wt = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    y = torch.empty(n, k, h, w, dtype=torch.bfloat16, device=dev).contiguous(memory_format=torch.channels_last)
    for _ in range(3):
        ext.conv_mm(x, wt, y)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(10):
        ext.conv_mm(x, wt, y)
    t1.record(); torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 100
    for _ in range(3):
        ym = F.conv2d(x, wt, padding=1)
    torch.cuda.synchronize()
    t0.record()
    for _ in range(10):
        ym = F.conv2d(x, wt, padding=1)
    t1.record(); torch.cuda.synchronize()
    us_m = t0.elapsed_time(t1) * 100
    gf = 2 * n * k * c * 9 * h * w / 1e9
    print(f"fwd N{n} C{c} {h}x{w} K{k}: ours {us:7.1f}us ({gf/us*1e3:6.0f} GF/s) miopen {us_m:7.1f}us ({gf/us_m*1e3:6.0f} GF/s)")

# stride-2 backward-data: polyphase MFMA kernel vs library (CK) path
for n, c, ho, wo, k in [(1024, 64, 16, 16, 128)]:
    hi, wi = 2 * ho, 2 * wo
    dy = torch.randn(n, k, ho, wo, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16)
    wt = w.permute(1, 2, 3, 0).contiguous()
    dx = torch.empty(n, c, hi, wi, dtype=torch.bfloat16, device=dev).contiguous(memory_format=torch.channels_last)
    assert ext.conv_s2_bwd_ok(n, ho, wo, k, c)
    for _ in range(3):
        ext.conv_s2_bwd(dy, wt, dx)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(10):
        ext.conv_s2_bwd(dy, wt, dx)
    t1.record(); torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 100
    wcl = w.contiguous(memory_format=torch.channels_last)
    xref = torch.randn(n, c, hi, wi, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    def lib():
        return torch.ops.aten.convolution_backward(
            dy, xref, wcl, None, [2, 2], [1, 1], [1, 1], False, [0, 0], 1,
            [True, False, False])[0]
    for _ in range(3):
        lib()
    torch.cuda.synchronize()
    t0.record()
    for _ in range(10):
        lib()
    t1.record(); torch.cuda.synchronize()
    us_m = t0.elapsed_time(t1) * 100
    gf = 2 * n * k * c * 9 * ho * wo / 1e9
    print(f"s2bwd N{n} K{k} {ho}x{wo}->C{c}: ours {us:7.1f}us ({gf/us*1e3:6.0f} GF/s) lib {us_m:7.1f}us ({gf/us_m*1e3:6.0f} GF/s)")

# stride-2 forward: experimental polyphase kernel vs library
for n, c, h, w_, k in [(1024, 64, 32, 32, 128)]:
    if not ext.conv_s2_fwd_ok(n, h, w_, c, k):
        print("s2fwd: shape unsupported"); break
    x = torch.randn(n, c, h, w_, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    y = torch.empty(n, k, h // 2, w_ // 2, dtype=torch.bfloat16, device=dev).contiguous(memory_format=torch.channels_last)
    for _ in range(3):
        ext.conv_s2_fwd(x, w, y)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(10):
        ext.conv_s2_fwd(x, w, y)
    t1.record(); torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 100
    for _ in range(3):
        ym = F.conv2d(x, w, stride=2, padding=1)
    torch.cuda.synchronize()
    t0.record()
    for _ in range(10):
        ym = F.conv2d(x, w, stride=2, padding=1)
    t1.record(); torch.cuda.synchronize()
    us_m = t0.elapsed_time(t1) * 100
    gf = 2 * n * k * c * 9 * (h // 2) * (w_ // 2) / 1e9
    print(f"s2fwd N{n} C{c} {h}x{w_}->K{k}: ours {us:7.1f}us ({gf/us*1e3:6.0f} GF/s) miopen {us_m:7.1f}us ({gf/us_m*1e3:6.0f} GF/s)")

# stride-2 weight gradient: experimental polyphase kernel vs library
for n, c, ho, wo, k in [(1024, 64, 16, 16, 128), (1024, 128, 8, 8, 256)]:
    if not ext.conv_s2_wrw_ok(n, ho, wo, c, k):
        print(f"s2wrw C{c}: shape unsupported"); continue
    hi, wi = 2 * ho, 2 * wo
    x = torch.randn(n, c, hi, wi, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    dy = torch.randn(n, k, ho, wo, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    nsplit = ext.conv_s2_wrw_nsplit(n, ho, wo, c, k)
    ws = torch.empty(nsplit * k * 9 * c, dtype=torch.float32, device=dev)
    dw = torch.empty(k, c, 3, 3, dtype=torch.float32, device=dev).contiguous(memory_format=torch.channels_last)
    for _ in range(3):
        ext.conv_s2_wrw(x, dy, ws, dw)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(10):
        ext.conv_s2_wrw(x, dy, ws, dw)
    t1.record(); torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 100
    wcl = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    def libw():
        return torch.ops.aten.convolution_backward(
            dy, x, wcl, None, [2, 2], [1, 1], [1, 1], False, [0, 0], 1,
            [False, True, False])[1]
    for _ in range(3):
        libw()
    torch.cuda.synchronize()
    t0.record()
    for _ in range(10):
        libw()
    t1.record(); torch.cuda.synchronize()
    us_m = t0.elapsed_time(t1) * 100
    gf = 2 * n * k * c * 9 * ho * wo / 1e9
    print(f"s2wrw N{n} {hi}x{wi}xC{c}->K{k}: ours {us:7.1f}us ({gf/us*1e3:6.0f} GF/s) lib {us_m:7.1f}us ({gf/us_m*1e3:6.0f} GF/s)")

# 1x1 stride-2 downsample: GEMM formulation vs library conv (fwd+bwd)
for n, c, hi, wi, k in [(1024, 64, 32, 32, 128)]:
    from adaptdl_amd.torch.layers import _S2Conv1x1Function
    x = torch.randn(n, c, hi, wi, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last).requires_grad_(True)
    wt1 = torch.randn(k, c, 1, 1, device=dev, requires_grad=True)
    def gemm_fb():
        y = _S2Conv1x1Function.apply(x, wt1)
        y.backward(torch.ones_like(y))
        x.grad = None; wt1.grad = None
    def lib_fb():
        y = F.conv2d(x, wt1.to(torch.bfloat16), stride=2)
        y.backward(torch.ones_like(y))
        x.grad = None; wt1.grad = None
    for fn, tag in ((gemm_fb, "gemm"), (lib_fb, "lib ")):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for _ in range(10):
            fn()
        t1.record(); torch.cuda.synchronize()
        print(f"s2 1x1 N{n} C{c}->K{k} fwd+bwd {tag}: {t0.elapsed_time(t1)*100:8.1f}us")

# Wo=8 s2 bwd-data: P2=4 redesign vs shipped w8 kernel vs library
for n, c, ho, wo, k in [(1024, 128, 8, 8, 256)]:
    if not ext.conv_s2_bwd_w8b_ok(n, ho, wo, k, c):
        print("w8b: shape unsupported"); break
    hi, wi = 2 * ho, 2 * wo
    dy = torch.randn(n, k, ho, wo, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(k, c, 3, 3, device=dev) * 0.1).to(torch.bfloat16)
    wt2 = w.permute(1, 2, 3, 0).contiguous()
    dx = torch.empty(n, c, hi, wi, dtype=torch.bfloat16, device=dev).contiguous(memory_format=torch.channels_last)
    gf = 2 * n * k * c * 9 * ho * wo / 1e9
    for fn, tag in ((lambda: ext.conv_s2_bwd_w8b(dy, wt2, dx), "w8b"),
                    (lambda: ext.conv_s2_bwd(dy, wt2, dx), "w8 ")):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for _ in range(10):
            fn()
        t1.record(); torch.cuda.synchronize()
        us = t0.elapsed_time(t1) * 100
        print(f"s2bwd-w8 N{n} K{k} {ho}x{wo}->C{c} {tag}: {us:7.1f}us ({gf/us*1e3:6.0f} GF/s)")
