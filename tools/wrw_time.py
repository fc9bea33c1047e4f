import torch, sys
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from adaptdl_amd import ops
ext = ops._load_extension()
dev = torch.device("cuda")
shapes = [(1024, 64, 32, 32, 64), (1024, 128, 16, 16, 128), (1024, 256, 8, 8, 256),
          # ImageNet-resolution (ResNet-50 at 224) padded-width shapes
          (256, 64, 56, 56, 64), (256, 128, 28, 28, 128),
          (256, 256, 14, 14, 256), (256, 512, 7, 7, 512)]
for n, c, h, w, k in shapes:
    x = torch.randn(n, c, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    dy = torch.randn(n, k, h, w, device=dev).to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    nsplit = ext.conv_wrw_nsplit(n, h, w, c, k)
    ws = torch.empty(nsplit * k * 9 * c, dtype=torch.float32, device=dev)
    dw = torch.empty(k, c, 3, 3, dtype=torch.float32, device=dev).contiguous(memory_format=torch.channels_last)
    for _ in range(3):
        ext.conv_wrw(x, dy, ws, dw)
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True); t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(10):
        ext.conv_wrw(x, dy, ws, dw)
    t1.record(); torch.cuda.synchronize()
    us = t0.elapsed_time(t1) * 100
    # MIOpen comparison
    x32 = x.float().requires_grad_(True)
    wt = torch.zeros(k, c, 3, 3, device=dev, dtype=torch.bfloat16).contiguous(memory_format=torch.channels_last).requires_grad_(True)
    import torch.nn.functional as F
    for _ in range(3):
        gw = torch.ops.aten.convolution_backward(dy, x, wt, None, [1,1], [1,1], [1,1], False, [0,0], 1, [False, True, False])[1]
    torch.cuda.synchronize()
    t0.record()
    for _ in range(10):
        gw = torch.ops.aten.convolution_backward(dy, x, wt, None, [1,1], [1,1], [1,1], False, [0,0], 1, [False, True, False])[1]
    t1.record(); torch.cuda.synchronize()
    us_m = t0.elapsed_time(t1) * 100
    gflop = 2 * n * k * c * 9 * h * w / 1e9
    print(f"shape N{n} C{c} {h}x{w} K{k}: ours {us:7.1f}us ({gflop/us*1e3:6.0f} GF/s)  miopen {us_m:7.1f}us ({gflop/us_m*1e3:6.0f} GF/s)  nsplit={nsplit}")
