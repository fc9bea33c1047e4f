"""Neural collaborative filtering with gradient accumulation.

Counterpart of /root/reference/examples/NCF/main.py: NeuMF trained with
implicit-feedback BCE and adaptive batch size with
gradient_accumulation=True (the reference's accumulation-path
workload).  Synthetic user/item interactions (no dataset downloads).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))


import argparse

import torch
import torch.nn.functional as F

import adaptdl_amd.env as env
import adaptdl_amd.torch as adl
from adaptdl_amd.models import NeuMF


def synthetic_interactions(num_users, num_items, n, neg_ratio=4):
    g = torch.Generator().manual_seed(9)
    users = torch.randint(0, num_users, (n,), generator=g)
    items = torch.randint(0, num_items, (n,), generator=g)
    labels = (torch.rand(n, generator=g) > neg_ratio /
              (neg_ratio + 1.0)).float()
    return torch.utils.data.TensorDataset(users, items, labels)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=4)
    parser.add_argument("--users", type=int, default=2000)
    parser.add_argument("--items", type=int, default=3000)
    parser.add_argument("--samples", type=int, default=100000)
    parser.add_argument("--bs", type=int, default=256)
    parser.add_argument("--max-bs", type=int, default=32768)
    parser.add_argument("--lr", type=float, default=1e-3)
    args = parser.parse_args()

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda" if use_gpu else "cpu")

    torch.manual_seed(33)
    model = NeuMF(args.users, args.items).to(device)
    optim = torch.optim.Adam(model.parameters(), lr=args.lr)
    adp = adl.AdaptiveDataParallel(model, optim)

    dataset = synthetic_interactions(args.users, args.items, args.samples)
    loader = adl.AdaptiveDataLoader(dataset, batch_size=args.bs,
                                    shuffle=True, drop_last=True)
    loader.autoscale_batch_size(args.max_bs,
                                local_bsz_bounds=(64, 8192),
                                gradient_accumulation=True)

    for epoch in adl.remaining_epochs_until(args.epochs):
        total, count = 0.0, 0
        for users, items, labels in loader:
            bsz = loader.current_batch_size
            users, items = users.to(device), items.to(device)
            labels = labels.to(device)
            optim.zero_grad()
            loss = F.binary_cross_entropy_with_logits(adp(users, items),
                                                      labels)
            loss.backward()
            optim.step()
            total += loss.item()
            count += 1
        if env.replica_rank() == 0 and count:
            print("epoch {} bce {:.4f} batch {} accum {}".format(
                epoch, total / count, bsz,
                loader.accumulation_steps))


if __name__ == "__main__":
    main()
