"""Per-step timing: raw torch loop vs ADP loop at bs 4096 and 1024."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F


def run_loop(label, step_fn, n=12):
    times = []
    for i in range(n):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        step_fn()
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) * 1000)
    print(label, " ".join("{:.0f}".format(t) for t in times), flush=True)


def main():
    from adaptdl_amd.models import ResNet18
    device = torch.device("cuda:0")
    torch.manual_seed(0)

    for bs in (1024, 4096):
        x = torch.randn(bs, 3, 32, 32, device=device)
        y = torch.randint(0, 10, (bs,), device=device)
        model = ResNet18().to(device)
        optim = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)

        def raw_step():
            optim.zero_grad(set_to_none=True)
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = F.cross_entropy(model(x), y)
            loss.backward()
            optim.step()

        run_loop("raw bs{}: ".format(bs), raw_step)

    # ADP arm (no dataloader; drive engine directly)
    import adaptdl_amd.collective as collective
    import adaptdl_amd.torch as adl
    if not collective.initialized():
        collective.initialize(master_addr="127.0.0.1")
    for bs in (1024, 4096):
        x = torch.randn(bs, 3, 32, 32, device=device)
        y = torch.randint(0, 10, (bs,), device=device)
        model = ResNet18().to(device)
        optim = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
        adp = adl.AdaptiveDataParallel(
            model, optim, name="adp-bs{}".format(bs))
        adp._after_sync = lambda: None  # no dataloader in this probe

        def adp_step():
            optim.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = F.cross_entropy(adp(x), y)
            loss.backward()
            adp.scaling_rule._orig_optimizer_step()

        run_loop("adp bs{}: ".format(bs), adp_step)

    # ADP arm minus GNS host math: engine only, stats kernels only
    from adaptdl_amd.torch.gradient_noise_scale import GradientNoiseScale

    class FakeADP:
        require_backward_grad_sync = True

        def _after_sync(self):
            pass

    bs = 4096
    x = torch.randn(bs, 3, 32, 32, device=device)
    y = torch.randint(0, 10, (bs,), device=device)
    model = ResNet18().to(device)
    optim = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9)
    gns = GradientNoiseScale(FakeADP(), optim, num_replicas=1)

    def gns_step():
        gns.reset_accumulation()
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loss = F.cross_entropy(model(x), y)
        loss.backward()
        optim.step()

    run_loop("gns-only bs4096: ", gns_step)


if __name__ == "__main__":
    main()
