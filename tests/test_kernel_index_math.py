"""CPU simulations of the HIP conv kernels' index math.

Each test mirrors the exact staging layout, tap decode, and store
mapping of a kernel in adaptdl_amd/ops/hip/conv_kernels.hip and checks
it against torch's reference op in fp32.  They run without a GPU, so a
refactor that breaks the polyphase index math fails here before it
costs GPU time (the bf16 numerics themselves are covered by the
@gpu tests in test_fused_conv.py).
"""

import torch


def test_s2_bwd_polyphase_decomposition():
    """k_conv3x3_s2_bwd: phase tap sets, dy window, store interleave."""
    torch.manual_seed(0)
    N, Ho, Wo, K, C, P2 = 2, 16, 16, 32, 32, 2
    Hi, Wi = 2 * Ho, 2 * Wo
    dy = torch.randn(N, Ho, Wo, K)
    w = torch.randn(K, C, 3, 3)
    ref = torch.nn.grad.conv2d_input(
        (N, C, Hi, Wi), w, dy.permute(0, 3, 1, 2), stride=2,
        padding=1).permute(0, 2, 3, 1)
    wt = w.permute(1, 2, 3, 0).reshape(C, 9, K)
    dx = torch.zeros(N, Hi, Wi, C)
    for n in range(N):
        for r0 in range(0, Ho, P2):
            win = torch.zeros(P2 + 1, Wo + 8, K)
            for j in range(P2 + 1):
                if r0 + j < Ho:
                    win[j, :Wo] = dy[n, r0 + j]
            stage = torch.zeros(4, P2, Wo, C)
            for ph in range(4):
                a, b = ph >> 1, ph & 1
                for h2 in range(P2):
                    for dh in ((0, 2) if a else (1,)):
                        for dw in ((0, 2) if b else (1,)):
                            j = h2 + (1 if dh == 0 else 0)
                            off = 1 if dw == 0 else 0
                            seg = win[j, off:off + Wo]
                            stage[ph, h2] += seg @ wt[:, dh * 3 + dw].t()
            for hl in range(2 * P2):
                for wi in range(Wi):
                    phc = ((hl & 1) << 1) | (wi & 1)
                    dx[n, 2 * r0 + hl, wi] = stage[phc, hl >> 1, wi >> 1]
    assert (dx - ref).abs().max().item() < 1e-3


def test_s2_fwd_eo_staging():
    """k_conv3x3_s2_fwd: E|z|O column-split staging + per-lane taps."""
    torch.manual_seed(2)
    N, H, W, C, K, P2 = 2, 16, 32, 16, 8, 4
    Ho, Wo = H // 2, W // 2
    x = torch.randn(N, H, W, C)
    w = torch.randn(K, C, 3, 3)
    ref = torch.nn.functional.conv2d(
        x.permute(0, 3, 1, 2), w, stride=2, padding=1).permute(0, 2, 3, 1)
    wt = w.permute(0, 2, 3, 1).reshape(K, 9, C)
    LS2 = 2 * Wo + 2
    y = torch.zeros(N, Ho, Wo, K)
    for n in range(N):
        for r0 in range(0, Ho, P2):
            xs = torch.zeros(2 * P2 + 1, LS2, C)
            for j in range(2 * P2 + 1):
                h = 2 * r0 - 1 + j
                if 0 <= h < H:
                    for wcol in range(W):
                        slot = (Wo + 1 + wcol // 2) if wcol % 2 \
                            else wcol // 2
                        xs[j, slot] = x[n, h, wcol]
            for ho_loc in range(P2):
                for wo in range(Wo):
                    acc = torch.zeros(K)
                    for dh in range(3):
                        for dw in range(3):
                            L = 2 * ho_loc + dh
                            slot = wo if dw == 1 else \
                                Wo + 1 + wo - (1 if dw == 0 else 0)
                            acc += wt[:, dh * 3 + dw] @ xs[L, slot]
                    y[n, r0 + ho_loc, wo] = acc
    assert (y - ref).abs().max().item() < 2e-3


def test_s2_wrw_polyphase_window():
    """k_conv3x3_s2_wrw: E|pad|O rows, window taps, dW accumulation."""
    torch.manual_seed(3)
    N, Ho, Wo, C, K, P = 2, 8, 8, 16, 8, 4
    Hi, Wi = 2 * Ho, 2 * Wo
    x = torch.randn(N, Hi, Wi, C)
    dy = torch.randn(N, Ho, Wo, K)
    xg = x.permute(0, 3, 1, 2).requires_grad_(True)
    w0 = torch.zeros(K, C, 3, 3, requires_grad=True)
    torch.nn.functional.conv2d(xg, w0, stride=2, padding=1) \
        .backward(dy.permute(0, 3, 1, 2))
    ref = w0.grad
    LS2 = 2 * Wo + 8
    dW = torch.zeros(K, 3, 3, C)
    for n in range(N):
        for lo0 in range(0, Ho, P):
            xt = torch.zeros(2 * P + 1, LS2, C)
            for j in range(2 * P + 1):
                h = 2 * lo0 - 1 + j
                if 0 <= h < Hi:
                    for wcol in range(Wi):
                        col = (Wo + 4 + (wcol - 1) // 2) if wcol % 2 \
                            else wcol // 2
                        xt[j, col] = x[n, h, wcol]
            for li in range(P):
                for wo in range(Wo):
                    for dh in range(3):
                        row = xt[2 * li + dh]
                        for dw in range(3):
                            if dw == 1:
                                v = row[wo]
                            else:
                                # O window: j = wo-1 (dw=0) or wo (dw=2);
                                # j = -1 hits the zeroed col Wo+3
                                j = wo - 1 if dw == 0 else wo
                                v = row[Wo + 4 + j]
                            dW[:, dh, dw] += torch.outer(
                                dy[n, lo0 + li, wo], v)
    err = (dW.permute(0, 3, 1, 2) - ref).abs().max().item()
    assert err < 1e-3, err


def test_w8b_wave_mapping():
    """k_conv3x3_s2_bwd_w8b: frag-quad coverage, one atomicAdd per
    k-quarter, invertible store mapping."""
    P2, Wo, CT = 4, 8, 16
    writes = {}
    for wid in range(8):
        wg2, kq = wid >> 2, wid & 3
        for fi in range(4):
            pha, phb = (0, 3) if wg2 == 0 else (1, 2)
            ph = phb if fi >> 1 else pha
            f = fi & 1
            for lane in range(64):
                for r in range(4):
                    m = (lane >> 4) * 4 + r
                    key = (ph, 2 * f + (m >> 3), m & 7, lane & 15)
                    writes.setdefault(key, []).append(kq)
    assert len(writes) == 4 * P2 * Wo * CT
    assert all(sorted(v) == [0, 1, 2, 3] for v in writes.values())
    for hl in range(2 * P2):
        for wi in range(2 * Wo):
            ph = ((hl & 1) << 1) | (wi & 1)
            s = (ph * P2 + (hl >> 1)) * Wo + (wi >> 1)
            ph2, rem = divmod(s, P2 * Wo)
            assert (ph2, rem // Wo, rem % Wo) == (ph, hl >> 1, wi >> 1)


def _wrw_params(W):
    """Mirror of conv3x3_wrw_params (conv_kernels.hip)."""
    table = {8: (8, 8), 16: (8, 16), 32: (4, 32),
             56: (1, 64), 28: (4, 32), 14: (8, 16), 7: (8, 8)}
    P, Wp = table[W]
    return P, Wp


def _simulate_wrw(x, dy, P, Wp):
    """CPU mirror of k_conv3x3_wrw's padded-width chunking: dy pixels
    with w >= W or h >= H stage zero; the tap read for dy pixel (li, w)
    and offset (dh, dw) is x_pad[li + dh][3 + w + dw] in the LS-padded
    line layout (4 left zeros)."""
    N, H, W, C = x.shape
    K = dy.shape[3]
    LS = Wp + 8
    lines = -(-H // P)
    dw_acc = torch.zeros(K, 9, C)
    for n in range(N):
        for chunk in range(lines):
            h0 = chunk * P
            x_pad = torch.zeros(P + 2, LS, C)
            for j in range(P + 2):
                h = h0 - 1 + j
                if 0 <= h < H:
                    x_pad[j, 4:4 + W] = x[n, h]
            dy_pad = torch.zeros(P, Wp, K)
            for li in range(P):
                h = h0 + li
                if h < H:
                    dy_pad[li, :W] = dy[n, h]
            for dh in range(3):
                for dwo in range(3):
                    tau = dh * 3 + dwo
                    # x cols for w = 0..Wp-1 at this tap
                    cols = torch.arange(Wp) + 3 + dwo
                    xt = x_pad[dh:dh + P, cols]     # (P, Wp, C)
                    dw_acc[:, tau] += torch.einsum(
                        "pwk,pwc->kc", dy_pad, xt)
    return dw_acc


def test_wrw_padded_width_chunking():
    """k_conv3x3_wrw generalized chunking vs torch conv2d_weight for the
    ImageNet-resolution widths (ResNet-50 at 224) and a legacy width."""
    torch.manual_seed(1)
    for (H, W, C, K, N) in [(7, 7, 64, 64, 2), (14, 14, 64, 64, 2),
                            (28, 28, 64, 64, 1), (56, 56, 64, 64, 1),
                            (16, 16, 64, 64, 2)]:
        P, Wp = _wrw_params(W)
        x = torch.randn(N, H, W, C)
        dy = torch.randn(N, H, W, K)
        got = _simulate_wrw(x, dy, P, Wp)
        ref = torch.nn.grad.conv2d_weight(
            x.permute(0, 3, 1, 2), (K, C, 3, 3),
            dy.permute(0, 3, 1, 2), stride=1, padding=1)
        ref = ref.permute(0, 2, 3, 1).reshape(K, 9, C)
        assert torch.allclose(got, ref, rtol=1e-4, atol=1e-3), (H, W)
