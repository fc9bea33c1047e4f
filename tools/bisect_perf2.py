"""Segment-level timing of the real bench loop (run on the GPU box)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

T = {}


def seg(name, t0):
    T[name] = T.get(name, 0.0) + (time.perf_counter() - t0)


def main():
    import adaptdl_amd.torch as adl
    from adaptdl_amd.models import ResNet18

    adl.init_process_group("nccl")
    device = torch.device("cuda:0")
    torch.manual_seed(0)
    model = ResNet18().to(device)
    optim = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)

    pool = 8192
    pool_x = torch.randn(pool, 3, 32, 32, device=device)
    pool_y = torch.randint(0, 10, (pool,), device=device)

    class SyntheticIndices(torch.utils.data.Dataset):
        def __len__(self):
            return 50000

        def __getitem__(self, i):
            return i

    loader = adl.AdaptiveDataLoader(SyntheticIndices(), batch_size=4096,
                                    collate_fn=lambda b: torch.as_tensor(b),
                                    num_workers=0)
    # Fixed total batch 4096 over 1 replica => local 4096, no accum,
    # comparable to the adaptive-chosen config (1024 x 4).
    n_steps = 0
    t_loop = None
    for epoch in adl.remaining_epochs_until(1):
        t0 = time.perf_counter()
        for idx in loader:
            seg("data", t0)
            t0 = time.perf_counter()
            idx = (idx % pool).to(device, non_blocking=True)
            x = pool_x[idx]
            y = pool_y[idx]
            seg("gather", t0)
            t0 = time.perf_counter()
            optim.zero_grad()
            seg("zero", t0)
            t0 = time.perf_counter()
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = F.cross_entropy(adp(x), y)
            seg("fwd", t0)
            t0 = time.perf_counter()
            loss.backward()
            seg("bwd", t0)
            t0 = time.perf_counter()
            optim.step()
            seg("opt", t0)
            n_steps += 1
            if n_steps == 25:
                break
            t0 = time.perf_counter()
        break
    torch.cuda.synchronize()
    print("steps:", n_steps)
    for k, v in sorted(T.items(), key=lambda kv: -kv[1]):
        print("{:10s} {:10.1f} ms total {:8.1f} ms/step".format(
            k, v * 1000, v * 1000 / n_steps))


if __name__ == "__main__":
    main()
