"""Transformer LM with adaptive BPTT batches (WikiText-2-class).

Counterpart of /root/reference/examples/transformer/transformer.py:
AdaptiveBPTTIterator with max_batch_size=1024*bs and local bounds
(16, 256) (reference transformer.py:157-170).  No dataset downloads in
this environment: a synthetic Zipf-distributed token stream stands in
for WikiText-2 (--tokens controls its length).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__)))))


import argparse
import faulthandler
import math
import signal

import torch

# Dump all thread stacks on SIGUSR1 (hung-collective debugging aid;
# reference transformer.py:34-35 parity).
faulthandler.register(signal.SIGUSR1)
import torch.nn.functional as F

import adaptdl_amd.env as env
import adaptdl_amd.torch as adl
from adaptdl_amd.models import TransformerLM


def synthetic_corpus(n_tokens, vocab):
    g = torch.Generator().manual_seed(11)
    weights = 1.0 / torch.arange(1, vocab + 1, dtype=torch.float)
    return torch.multinomial(weights, n_tokens, replacement=True,
                             generator=g)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=5)
    parser.add_argument("--bs", type=int, default=20)
    parser.add_argument("--bptt", type=int, default=35)
    parser.add_argument("--vocab", type=int, default=2000)
    parser.add_argument("--tokens", type=int, default=200000)
    parser.add_argument("--lr", type=float, default=5.0)
    args = parser.parse_args()

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda" if use_gpu else "cpu")

    torch.manual_seed(5)
    corpus = synthetic_corpus(args.tokens, args.vocab)
    model = TransformerLM(args.vocab).to(device)
    optim = torch.optim.SGD(model.parameters(), lr=args.lr)
    adp = adl.AdaptiveDataParallel(model, optim)

    it = adl.AdaptiveBPTTIterator(corpus, batch_size=args.bs,
                                  bptt_len=args.bptt,
                                  max_batch_size=args.bs * 64,
                                  local_bsz_bounds=(16, 256),
                                  # Extension over the reference: lets
                                  # a single replica scale its batch
                                  # via accumulation (BASELINE config 3)
                                  gradient_accumulation=True,
                                  device=device)
    for epoch in adl.remaining_epochs_until(args.epochs):
        total_loss, total_tok = 0.0, 0
        for text, target in it:
            optim.zero_grad()
            if use_gpu:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    out = adp(text)
                    loss = F.cross_entropy(
                        out.view(-1, args.vocab), target.reshape(-1))
            else:
                out = adp(text)
                loss = F.cross_entropy(out.view(-1, args.vocab),
                                       target.reshape(-1))
            loss.backward()
            torch.nn.utils.clip_grad_norm_(model.parameters(), 0.5)
            optim.step()
            total_loss += loss.item() * target.numel()
            total_tok += target.numel()
        if env.replica_rank() == 0 and total_tok:
            print("epoch {} ppl {:.2f}".format(
                epoch, math.exp(total_loss / total_tok)))


if __name__ == "__main__":
    main()
