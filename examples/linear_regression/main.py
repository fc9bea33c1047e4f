"""Minimal elastic workload: linear regression on CPU/gloo.

Counterpart of /root/reference/examples/linear_regression/main.py —
the 4-line-migration demo: init_process_group, AdaptiveDataParallel,
AdaptiveDataLoader, remaining_epochs_until, Accumulator.

Run standalone:           python main.py
Run elastically (2 way):  ADAPTDL_NUM_REPLICAS=2 ADAPTDL_REPLICA_RANK=r \
                          ADAPTDL_MASTER_PORT=p python main.py
or submit via the local controller:  adaptdl-amd run -- python main.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))


import argparse

import torch

import adaptdl_amd.torch as adl


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=30)
    parser.add_argument("--batch-size", type=int, default=16)
    parser.add_argument("--autoscale", action="store_true")
    args = parser.parse_args()

    adl.init_process_group(
        "nccl" if torch.cuda.is_available() else "gloo")

    torch.manual_seed(42)
    true_w = torch.tensor([[3.0], [4.0]])
    xs = torch.randn(1024, 2)
    ys = xs @ true_w + 0.01 * torch.randn(1024, 1)
    dataset = torch.utils.data.TensorDataset(xs, ys)

    model = torch.nn.Linear(2, 1, bias=False)
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    adp = adl.AdaptiveDataParallel(model, optim)

    loader = adl.AdaptiveDataLoader(dataset, batch_size=args.batch_size,
                                    shuffle=True)
    if args.autoscale:
        loader.autoscale_batch_size(256, local_bsz_bounds=(8, 128))

    stats = adl.Accumulator()
    for epoch in adl.remaining_epochs_until(args.epochs):
        for x, y in loader:
            optim.zero_grad()
            loss = ((adp(x) - y) ** 2).mean()
            loss.backward()
            optim.step()
            stats["loss_sum"] += loss.item() * len(x)
            stats["count"] += len(x)
        with stats.synchronized():
            print("epoch {} loss {:.5f}".format(
                epoch, stats["loss_sum"] / stats["count"]))
            stats["loss_sum"] = stats["count"] = 0
    print("learned:", model.weight.detach().reshape(-1).tolist())


if __name__ == "__main__":
    main()
