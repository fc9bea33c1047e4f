"""ResNet-18 CIFAR-10 with adaptive batch size on MI355X.

Counterpart of /root/reference/examples/pytorch-cifar/main.py: the
flagship adaptive-batch-size workload — autoscale_batch_size(4096,
(32, 1024), gradient_accumulation) (reference main.py:77), bf16
autocast, channels_last + the CDNA4 fused-BN ResNets of
adaptdl_amd.models.  There is no dataset download in this environment,
so --synthetic (default) trains on CIFAR-shaped random data; pass
--data DIR to use a torchvision-style CIFAR-10 folder if present.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))


import argparse
import os

import torch
import torch.nn.functional as F

import adaptdl_amd.env as env
import adaptdl_amd.torch as adl
from adaptdl_amd.models import ResNet18
from adaptdl_amd.models.cifar import CIFAR_MODELS


class SyntheticCifar(torch.utils.data.Dataset):
    def __init__(self, n=50000):
        g = torch.Generator().manual_seed(0)
        self.x = torch.randn(n, 3, 32, 32, generator=g)
        self.y = torch.randint(0, 10, (n,), generator=g)

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i], self.y[i]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=60)
    parser.add_argument("--bs", type=int, default=128)
    parser.add_argument("--max-bs", type=int, default=4096)
    parser.add_argument("--lr", type=float, default=0.1)
    parser.add_argument("--model", default="resnet18",
                        choices=["resnet18"] + sorted(CIFAR_MODELS),
                        help="architecture from the CIFAR model zoo")
    parser.add_argument("--synthetic", action="store_true", default=True)
    parser.add_argument("--samples", type=int, default=50000)
    args = parser.parse_args()

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda" if use_gpu else "cpu")

    torch.manual_seed(1234)
    ctor = ResNet18 if args.model == "resnet18" \
        else CIFAR_MODELS[args.model]
    model = ctor().to(device)
    if use_gpu:
        model = model.to(memory_format=torch.channels_last)
    optim = adl.FusedSGD(model.parameters(), lr=args.lr,
                         momentum=0.9, weight_decay=5e-4)
    sched = torch.optim.lr_scheduler.CosineAnnealingLR(optim, args.epochs)
    adp = adl.AdaptiveDataParallel(model, optim, sched)

    loader = adl.AdaptiveDataLoader(SyntheticCifar(args.samples),
                                    batch_size=args.bs, shuffle=True,
                                    drop_last=True)
    if args.max_bs > 0:   # --max-bs 0 disables adaptive batch sizing
        loader.autoscale_batch_size(args.max_bs,
                                    local_bsz_bounds=(32, 1024),
                                    gradient_accumulation=True)

    def fwd_bwd(x, y):
        optim.zero_grad()
        if use_gpu:
            with torch.autocast("cuda", dtype=torch.bfloat16):
                loss = F.cross_entropy(adp(x), y)
        else:
            loss = F.cross_entropy(adp(x), y)
        loss.backward()
        return loss

    # Experimental: ADAPTDL_HIPGRAPH=1 captures the steady microbatch
    # cycle into hipGraphs (see adaptdl_amd/torch/graph_step.py).
    from adaptdl_amd.torch.graph_step import maybe_graphed_stepper
    stepper = maybe_graphed_stepper(adp, optim, fwd_bwd)

    stats = adl.Accumulator()
    for epoch in adl.remaining_epochs_until(args.epochs):
        model.train()
        bsz = None
        for x, y in loader:
            bsz = loader.current_batch_size
            x, y = x.to(device), y.to(device)
            if use_gpu:
                x = x.contiguous(memory_format=torch.channels_last)
            if stepper is not None:
                loss = stepper.microbatch(x, y)
            else:
                loss = fwd_bwd(x, y)
            optim.step()
            if loss is not None:
                stats["loss_sum"] += loss.item() * len(y)
                stats["count"] += len(y)
        sched.step()
        with stats.synchronized():
            if int(os.getenv("ADAPTDL_REPLICA_RANK", "0")) == 0:
                print("epoch {} replicas {} batch {} loss {:.4f}".format(
                    epoch, env.num_replicas(), bsz,
                    stats["loss_sum"] / max(stats["count"], 1)))
            stats["loss_sum"] = stats["count"] = 0


if __name__ == "__main__":
    main()
