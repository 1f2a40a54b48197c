"""DCGAN-style multi-model / multi-loss amp example (reference:
examples/dcgan/main_amp.py:214-253 — the surface that pins
``amp.initialize([netD, netG], [optD, optG], num_losses=3)`` and
``amp.scale_loss(err, opt, loss_id=k)``). Synthetic-noise edition."""

import argparse

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), "..", ".."))

import torch
import torch.nn as nn


class Generator(nn.Module):
    def __init__(self, nz=100, ngf=32, nc=3):
        super().__init__()
        self.main = nn.Sequential(
            nn.ConvTranspose2d(nz, ngf * 4, 4, 1, 0, bias=False),
            nn.BatchNorm2d(ngf * 4), nn.ReLU(True),
            nn.ConvTranspose2d(ngf * 4, ngf * 2, 4, 2, 1, bias=False),
            nn.BatchNorm2d(ngf * 2), nn.ReLU(True),
            nn.ConvTranspose2d(ngf * 2, ngf, 4, 2, 1, bias=False),
            nn.BatchNorm2d(ngf), nn.ReLU(True),
            nn.ConvTranspose2d(ngf, nc, 4, 2, 1, bias=False),
            nn.Tanh(),
        )

    def forward(self, z):
        return self.main(z)


class Discriminator(nn.Module):
    def __init__(self, ndf=32, nc=3):
        super().__init__()
        self.main = nn.Sequential(
            nn.Conv2d(nc, ndf, 4, 2, 1, bias=False), nn.LeakyReLU(0.2, True),
            nn.Conv2d(ndf, ndf * 2, 4, 2, 1, bias=False),
            nn.BatchNorm2d(ndf * 2), nn.LeakyReLU(0.2, True),
            nn.Conv2d(ndf * 2, ndf * 4, 4, 2, 1, bias=False),
            nn.BatchNorm2d(ndf * 4), nn.LeakyReLU(0.2, True),
            nn.Conv2d(ndf * 4, 1, 4, 1, 0, bias=False),
        )

    def forward(self, x):
        return self.main(x).reshape(-1)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--batch-size", type=int, default=32)
    ap.add_argument("--opt-level", default="O1")
    args = ap.parse_args()

    from apex_amd import amp
    from apex_amd.optimizers import FusedAdam

    device = "cuda" if torch.cuda.is_available() else "cpu"
    torch.manual_seed(0)
    nz = 100
    netG = Generator(nz).to(device)
    netD = Discriminator().to(device)
    optD = FusedAdam(netD.parameters(), lr=2e-4, betas=(0.5, 0.999), adam_w_mode=False)
    optG = FusedAdam(netG.parameters(), lr=2e-4, betas=(0.5, 0.999), adam_w_mode=False)

    # multi-model, multi-loss initialization (3 losses: D-real, D-fake, G)
    [netD, netG], [optD, optG] = amp.initialize(
        [netD, netG], [optD, optG], opt_level=args.opt_level,
        cast_model_type=None if args.opt_level in ("O0", "O1") else torch.bfloat16,
        num_losses=3, verbosity=0,
    )
    criterion = nn.BCEWithLogitsLoss()

    for it in range(args.iters):
        real = torch.randn(args.batch_size, 3, 32, 32, device=device)
        noise = torch.randn(args.batch_size, nz, 1, 1, device=device)
        real_label = torch.ones(args.batch_size, device=device)
        fake_label = torch.zeros(args.batch_size, device=device)

        # --- D step ---
        optD.zero_grad()
        errD_real = criterion(netD(real).float(), real_label)
        with amp.scale_loss(errD_real, optD, loss_id=0) as errD_real_scaled:
            errD_real_scaled.backward()
        fake = netG(noise)
        errD_fake = criterion(netD(fake.detach()).float(), fake_label)
        with amp.scale_loss(errD_fake, optD, loss_id=1) as errD_fake_scaled:
            errD_fake_scaled.backward()
        optD.step()

        # --- G step ---
        optG.zero_grad()
        errG = criterion(netD(fake).float(), real_label)
        with amp.scale_loss(errG, optG, loss_id=2) as errG_scaled:
            errG_scaled.backward()
        optG.step()

        if it % 5 == 0:
            print(f"iter {it:3d}  errD {float(errD_real + errD_fake):.4f}  errG {float(errG):.4f}")


if __name__ == "__main__":
    main()
