"""BERT pretraining with AdamScale adaptive batches.

Counterpart of /root/reference/examples/BERT/mlm_task_adaptdl.py
(--task mlm, default) and ns_task_adaptdl.py (--task ns: sentence-pair
next-sentence classification through BertForPreTraining): Adam
optimizer (so AdaptiveDataParallel selects AdamScale +
AdamGradientNoiseScale), adaptive batch size with accumulation.
Synthetic token data stands in for the corpus (no network in this
environment); --config mini keeps CPU runs fast, base matches the
8-GPU workload.
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
from adaptdl_amd.models import (BertConfig, BertForMaskedLM,
                                BertForPreTraining)

MASK_ID = 4
PAD_ID = 0


class SyntheticNS(torch.utils.data.Dataset):
    """Sentence pairs; label 1 iff the second segment follows the first
    (synthetic stand-in for the reference ns_task corpus): "sentences"
    are runs from a fixed token stream, negatives are random jumps."""

    def __init__(self, config, n=8192, seq_len=128):
        g = torch.Generator().manual_seed(5)
        self.stream = torch.randint(10, config.vocab_size, (n * 8,),
                                    generator=g)
        self.n, self.seq_len = n, seq_len
        self.g = g

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        half = self.seq_len // 2
        a0 = (i * 7) % (len(self.stream) - self.seq_len)
        seg_a = self.stream[a0:a0 + half]
        follows = i % 2 == 0
        if follows:
            seg_b = self.stream[a0 + half:a0 + 2 * half]
        else:
            b0 = ((i * 31 + 13) * half) % (len(self.stream) - half)
            seg_b = self.stream[b0:b0 + half]
        tokens = torch.cat([seg_a, seg_b])
        type_ids = torch.cat([torch.zeros(half, dtype=torch.long),
                              torch.ones(half, dtype=torch.long)])
        return tokens, type_ids, torch.tensor(int(follows))


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
    parser.add_argument("--task", choices=["mlm", "ns"], default="mlm")
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
    if args.task == "ns":
        model = BertForPreTraining(config).to(device)
        dataset = SyntheticNS(config, args.samples, args.seq_len)
    else:
        model = BertForMaskedLM(config).to(device)
        dataset = SyntheticMLM(config, args.samples, args.seq_len)
    optim = adl.FusedAdam(model.parameters(), lr=args.lr)
    adp = adl.AdaptiveDataParallel(model, optim)  # -> AdamScale

    loader = adl.AdaptiveDataLoader(
        dataset, batch_size=args.bs, shuffle=True, drop_last=True)
    loader.autoscale_batch_size(args.max_bs, local_bsz_bounds=(8, 256),
                                gradient_accumulation=True)

    def ns_loss(batch):
        tokens, type_ids, label = (t.to(device) for t in batch)
        _, nsp_logits = adp(tokens, token_type_ids=type_ids)
        return F.cross_entropy(nsp_logits, label)

    for epoch in adl.remaining_epochs_until(args.epochs):
        total, count = 0.0, 0
        bsz = None
        if args.task == "ns":
            for batch in loader:
                bsz = loader.current_batch_size
                optim.zero_grad()
                if use_gpu:
                    with torch.autocast("cuda", dtype=torch.bfloat16):
                        loss = ns_loss(batch)
                else:
                    loss = ns_loss(batch)
                loss.backward()
                optim.step()
                total += loss.item()
                count += 1
            if env.replica_rank() == 0 and count:
                print("epoch {} ns-loss {:.4f} batch {}".format(
                    epoch, total / count, bsz))
            continue
        for tokens, labels in loader:
            bsz = loader.current_batch_size
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
                epoch, total / count, bsz))


if __name__ == "__main__":
    main()
