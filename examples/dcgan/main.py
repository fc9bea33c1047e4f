"""DCGAN: two AdaptiveDataParallel instances ("netD", "netG").

Counterpart of /root/reference/examples/dcgan/dcgan.py (which registers
two named ADP states, dcgan.py:500-501): generator and discriminator
each wrapped in their own named AdaptiveDataParallel so both are
checkpointed/restored and contribute grad params independently.
Synthetic 64x64 image data (no dataset downloads).
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
from adaptdl_amd.models import Generator, Discriminator


class SyntheticImages(torch.utils.data.Dataset):
    def __init__(self, n=4096):
        g = torch.Generator().manual_seed(17)
        self.x = torch.tanh(torch.randn(n, 3, 64, 64, generator=g))

    def __len__(self):
        return len(self.x)

    def __getitem__(self, i):
        return self.x[i]


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=3)
    parser.add_argument("--bs", type=int, default=64)
    parser.add_argument("--nz", type=int, default=100)
    parser.add_argument("--samples", type=int, default=4096)
    parser.add_argument("--lr", type=float, default=2e-4)
    args = parser.parse_args()

    use_gpu = torch.cuda.is_available()
    adl.init_process_group("nccl" if use_gpu else "gloo")
    device = torch.device("cuda" if use_gpu else "cpu")

    torch.manual_seed(99)
    netG = Generator(nz=args.nz).to(device)
    netD = Discriminator().to(device)
    optG = torch.optim.Adam(netG.parameters(), lr=args.lr,
                            betas=(0.5, 0.999))
    optD = torch.optim.Adam(netD.parameters(), lr=args.lr,
                            betas=(0.5, 0.999))
    adpG = adl.AdaptiveDataParallel(netG, optG, name="netG")
    adpD = adl.AdaptiveDataParallel(netD, optD, name="netD")

    loader = adl.AdaptiveDataLoader(SyntheticImages(args.samples),
                                    batch_size=args.bs, shuffle=True,
                                    drop_last=True)

    for epoch in adl.remaining_epochs_until(args.epochs):
        d_sum = g_sum = 0.0
        batches = 0
        for real in loader:
            real = real.to(device)
            b = real.size(0)
            ones = torch.ones(b, device=device)
            zeros = torch.zeros(b, device=device)
            # --- discriminator step ---
            optD.zero_grad()
            noise = torch.randn(b, args.nz, 1, 1, device=device)
            fake = adpG(noise)
            loss_d = F.binary_cross_entropy_with_logits(adpD(real), ones) \
                + F.binary_cross_entropy_with_logits(
                    adpD(fake.detach()), zeros)
            loss_d.backward()
            optD.step()
            # --- generator step ---
            optG.zero_grad()
            loss_g = F.binary_cross_entropy_with_logits(adpD(fake), ones)
            loss_g.backward()
            optG.step()
            d_sum += loss_d.item()
            g_sum += loss_g.item()
            batches += 1
        if env.replica_rank() == 0 and batches:
            print("epoch {} lossD {:.3f} lossG {:.3f}".format(
                epoch, d_sum / batches, g_sum / batches))


if __name__ == "__main__":
    main()
