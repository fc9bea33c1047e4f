"""DCGAN generator/discriminator — the two-ADP-instances workload.

Self-contained counterpart of the reference's DCGAN example
(/root/reference/examples/dcgan/dcgan.py, which wraps netD and netG in
two named AdaptiveDataParallel instances, dcgan.py:500-501).  Standard
64x64 DCGAN topology; BatchNorm uses the framework's
FusedBatchNormAct2d so the generator's BN+ReLU pairs run the CDNA4
fused kernels on MI355X (LeakyReLU in the discriminator stays separate).
"""

import torch.nn as nn

from adaptdl_amd.torch.layers import FusedBatchNormAct2d


class Generator(nn.Module):
    def __init__(self, nz=100, ngf=64, nc=3):
        super().__init__()
        self.main = nn.Sequential(
            nn.ConvTranspose2d(nz, ngf * 8, 4, 1, 0, bias=False),
            FusedBatchNormAct2d(ngf * 8, relu=True),
            nn.ConvTranspose2d(ngf * 8, ngf * 4, 4, 2, 1, bias=False),
            FusedBatchNormAct2d(ngf * 4, relu=True),
            nn.ConvTranspose2d(ngf * 4, ngf * 2, 4, 2, 1, bias=False),
            FusedBatchNormAct2d(ngf * 2, relu=True),
            nn.ConvTranspose2d(ngf * 2, ngf, 4, 2, 1, bias=False),
            FusedBatchNormAct2d(ngf, relu=True),
            nn.ConvTranspose2d(ngf, nc, 4, 2, 1, bias=False),
            nn.Tanh())

    def forward(self, z):
        return self.main(z)


class Discriminator(nn.Module):
    def __init__(self, ndf=64, nc=3):
        super().__init__()
        self.main = nn.Sequential(
            nn.Conv2d(nc, ndf, 4, 2, 1, bias=False),
            nn.LeakyReLU(0.2, inplace=True),
            nn.Conv2d(ndf, ndf * 2, 4, 2, 1, bias=False),
            FusedBatchNormAct2d(ndf * 2),
            nn.LeakyReLU(0.2, inplace=True),
            nn.Conv2d(ndf * 2, ndf * 4, 4, 2, 1, bias=False),
            FusedBatchNormAct2d(ndf * 4),
            nn.LeakyReLU(0.2, inplace=True),
            nn.Conv2d(ndf * 4, ndf * 8, 4, 2, 1, bias=False),
            FusedBatchNormAct2d(ndf * 8),
            nn.LeakyReLU(0.2, inplace=True),
            nn.Conv2d(ndf * 8, 1, 4, 1, 0, bias=False),
            nn.Flatten())

    def forward(self, x):
        return self.main(x).squeeze(-1)
