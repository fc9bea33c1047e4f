"""BERT masked-LM pretraining with AdamScale adaptive batches.

Counterpart of /root/reference/examples/BERT/mlm_task_adaptdl.py: Adam
optimizer (so AdaptiveDataParallel selects AdamScale +
AdamGradientNoiseScale), adaptive batch size with accumulation, custom
dataloader built on AdaptiveDataLoaderMixin semantics.  Synthetic token
data stands in for the corpus (no network in this environment);
--config mini keeps CPU runs fast, base matches the 8-GPU workload.
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
from adaptdl_amd.models import BertConfig, BertForMaskedLM

MASK_ID = 4
PAD_ID = 0


class SyntheticMLM(torch.utils.data.Dataset):
    def __init__(self, config, n=8192, seq_len=128):
        g = torch.Generator().manual_seed(3)
        self.tokens = torch.randint(10, config.vocab_size, (n, seq_len),
                                    generator=g)
        self.g = torch.Generator().manual_seed(4)

    def __len__(self):
        return len(self.tokens)

    def __getitem__(self, i):
        tokens = self.tokens[i]
        labels = torch.full_like(tokens, -100)
        mask = torch.rand(tokens.shape, generator=self.g) < 0.15
        labels[mask] = tokens[mask]
        corrupted = tokens.clone()
        corrupted[mask] = MASK_ID
        return corrupted, labels


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--config", choices=["mini", "base"],
                        default="mini")
    parser.add_argument("--epochs", type=int, default=3)
    parser.add_argument("--bs", type=int, default=32)
    parser.add_argument("--max-bs", type=int, default=1024)
    parser.add_argument("--seq-len", type=int, default=128)
    parser.add_argument("--samples", type=int, default=8192)
    parser.add_argument("--lr", type=float, default=1e-4)
    args = parser.parse_args()

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda" if use_gpu else "cpu")

    config = (BertConfig.mini() if args.config == "mini"
              else BertConfig.base())
    torch.manual_seed(21)
    model = BertForMaskedLM(config).to(device)
    optim = adl.FusedAdam(model.parameters(), lr=args.lr)
    adp = adl.AdaptiveDataParallel(model, optim)  # -> AdamScale

    loader = adl.AdaptiveDataLoader(
        SyntheticMLM(config, args.samples, args.seq_len),
        batch_size=args.bs, shuffle=True, drop_last=True)
    loader.autoscale_batch_size(args.max_bs, local_bsz_bounds=(8, 256),
                                gradient_accumulation=True)

    for epoch in adl.remaining_epochs_until(args.epochs):
        total, count = 0.0, 0
        for tokens, labels in loader:
            tokens, labels = tokens.to(device), labels.to(device)
            optim.zero_grad()
            if use_gpu:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    logits = adp(tokens)
                    loss = F.cross_entropy(
                        logits.view(-1, config.vocab_size),
                        labels.view(-1), ignore_index=-100)
            else:
                logits = adp(tokens)
                loss = F.cross_entropy(logits.view(-1, config.vocab_size),
                                       labels.view(-1), ignore_index=-100)
            loss.backward()
            optim.step()
            total += loss.item()
            count += 1
        if env.replica_rank() == 0 and count:
            print("epoch {} mlm-loss {:.4f} batch {}".format(
                epoch, total / count, loader.current_batch_size))


if __name__ == "__main__":
    main()
