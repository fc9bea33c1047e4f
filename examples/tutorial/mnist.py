"""The canonical 4-line migration tutorial (MNIST-shaped).

Counterpart of /root/reference/tutorial/mnist_step_5.py — the finished
tutorial program after all four AdaptDL changes:
  1. adl.init_process_group(...)
  2. AdaptiveDataParallel(model, optimizer)
  3. AdaptiveDataLoader(...) [+ autoscale_batch_size]
  4. remaining_epochs_until(...) + Accumulator for restart-safe stats

No dataset downloads in this environment: MNIST-shaped synthetic data
(28x28 grayscale, 10 classes) with a deterministic teacher so accuracy
is meaningful and restart-identical.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))


import argparse

import torch
import torch.nn as nn
import torch.nn.functional as F

import adaptdl_amd.env as env
import adaptdl_amd.torch as adl


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, 3, 1)
        self.conv2 = nn.Conv2d(32, 64, 3, 1)
        self.fc1 = nn.Linear(9216, 128)
        self.fc2 = nn.Linear(128, 10)

    def forward(self, x):
        x = F.relu(self.conv1(x))
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = torch.flatten(x, 1)
        return self.fc2(F.relu(self.fc1(x)))


def synthetic_mnist(n, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, 1, 28, 28, generator=g)
    teacher = torch.randn(28 * 28, 10, generator=g)
    y = (x.view(n, -1) @ teacher).argmax(1)
    return torch.utils.data.TensorDataset(x, y)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=5)
    parser.add_argument("--bs", type=int, default=64)
    parser.add_argument("--lr", type=float, default=0.05)
    parser.add_argument("--train-samples", type=int, default=8192)
    parser.add_argument("--test-samples", type=int, default=1024)
    args = parser.parse_args()

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")        # (1)
    device = torch.device("cuda" if use_gpu else "cpu")

    torch.manual_seed(1)
    model = Net().to(device)
    optim = torch.optim.SGD(model.parameters(), lr=args.lr,
                            momentum=0.9)
    adp = adl.AdaptiveDataParallel(model, optim)                 # (2)

    train_loader = adl.AdaptiveDataLoader(                       # (3)
        synthetic_mnist(args.train_samples, 0),
        batch_size=args.bs, shuffle=True, drop_last=True)
    train_loader.autoscale_batch_size(1024,
                                      local_bsz_bounds=(16, 256))
    test_loader = adl.AdaptiveDataLoader(
        synthetic_mnist(args.test_samples, 1), batch_size=256)

    for epoch in adl.remaining_epochs_until(args.epochs):        # (4)
        model.train()
        for x, y in train_loader:
            x, y = x.to(device), y.to(device)
            optim.zero_grad()
            F.nll_loss(F.log_softmax(adp(x), 1), y).backward()
            optim.step()
        model.eval()
        stats = adl.Accumulator()
        with torch.no_grad():
            for x, y in test_loader:
                x, y = x.to(device), y.to(device)
                pred = adp(x).argmax(1)
                stats["correct"] += (pred == y).sum().item()
                stats["total"] += len(y)
        with stats.synchronized():
            if env.replica_rank() == 0:
                print("epoch {}: test acc {}/{} = {:.3f}".format(
                    epoch, int(stats["correct"]), int(stats["total"]),
                    stats["correct"] / stats["total"]))


if __name__ == "__main__":
    main()
