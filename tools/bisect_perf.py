"""Host-overhead bisection for the bench loop (run on the GPU box)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def timeit(label, fn, steps=10):
    fn()  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps * 1000
    print("{:40s} {:10.1f} ms/step".format(label, dt), flush=True)
    return dt


def main():
    from adaptdl_amd.models import ResNet18
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    bs = 1024

    x = torch.randn(bs, 3, 32, 32, device=device)
    y = torch.randint(0, 10, (bs,), device=device)

    # (a) raw loop, plain SGD, no adaptdl
    model = ResNet18().to(device)
    optim = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)

    def raw_step():
        optim.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        optim.step()

    timeit("raw bf16 bs1024", raw_step)

    # (a2) with cudnn benchmark
    torch.backends.cudnn.benchmark = True
    timeit("raw bf16 bs1024 + cudnn.benchmark", raw_step)

    # (a3) channels_last
    model_cl = ResNet18().to(device).to(memory_format=torch.channels_last)
    optim_cl = torch.optim.SGD(model_cl.parameters(), lr=0.1, momentum=0.9)
    x_cl = x.to(memory_format=torch.channels_last)

    def cl_step():
        optim_cl.zero_grad(set_to_none=True)
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model_cl(x_cl), y)
        loss.backward()
        optim_cl.step()

    timeit("raw bf16 bs1024 channels_last", cl_step)

    # (b) ADP-wrapped, single process, no dataloader
    import adaptdl_amd.torch as adl
    import adaptdl_amd.collective as collective
    if not collective.initialized():
        collective.initialize(master_addr="127.0.0.1")
    model2 = ResNet18().to(device)
    optim2 = torch.optim.SGD(model2.parameters(), lr=0.1, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model2, optim2)

    class _FakeDL:
        pass

    # Drive without dataloader: require_sync always True.
    def adp_step():
        optim2.zero_grad()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(adp.module(x), y)
        loss.backward()
        # bypass scaling-rule step (needs dataloader); raw optimizer step:
        adp.scaling_rule._orig_optimizer_step()

    timeit("ADP engine (hooks+stats), no loader", adp_step)

    # (c) dataloader iteration cost alone
    dataset_len = 50000

    class SyntheticIndices(torch.utils.data.Dataset):
        def __len__(self):
            return dataset_len

        def __getitem__(self, i):
            return i

    loader = torch.utils.data.DataLoader(
        SyntheticIndices(), batch_size=bs,
        collate_fn=lambda b: torch.as_tensor(b))
    it = iter(loader)

    def next_batch():
        nonlocal it
        try:
            next(it)
        except StopIteration:
            it = iter(loader)
            next(it)

    timeit("plain DataLoader next(bs=1024)", next_batch, steps=20)

    # (d) scipy fit cost (runs rank-0 every 30 s)
    from adaptdl_amd.goodput import fit_perf_params
    import numpy as np
    nn_ = np.ones(8)
    rr = np.array([1, 1, 1, 1, 1, 1, 1, 1])
    bb = np.array([128, 128, 256, 256, 512, 512, 1024, 1024])
    at = 0.1 + 0.0001 * bb
    ot = at + 0.01
    t0 = time.perf_counter()
    fit_perf_params(nn_, rr, bb, at, ot)
    print("{:40s} {:10.1f} ms".format("fit_perf_params once",
                                      (time.perf_counter() - t0) * 1000))


if __name__ == "__main__":
    main()
