"""CycleGAN generators/discriminators for the cyclegan workload
(reference workloads/pytorch/cyclegan: ResNet-based generator + PatchGAN)."""

from __future__ import annotations

import torch
import torch.nn as nn


class ResidualBlock(nn.Module):
    def __init__(self, ch):
        super().__init__()
        self.block = nn.Sequential(
            nn.ReflectionPad2d(1),
            nn.Conv2d(ch, ch, 3),
            nn.InstanceNorm2d(ch),
            nn.ReLU(inplace=True),
            nn.ReflectionPad2d(1),
            nn.Conv2d(ch, ch, 3),
            nn.InstanceNorm2d(ch),
        )

    def forward(self, x):
        return x + self.block(x)


class GeneratorResNet(nn.Module):
    def __init__(self, channels=3, num_residual_blocks=9, base=64):
        super().__init__()
        layers = [
            nn.ReflectionPad2d(3),
            nn.Conv2d(channels, base, 7),
            nn.InstanceNorm2d(base),
            nn.ReLU(inplace=True),
        ]
        ch = base
        for _ in range(2):  # downsample
            layers += [
                nn.Conv2d(ch, ch * 2, 3, 2, 1),
                nn.InstanceNorm2d(ch * 2),
                nn.ReLU(inplace=True),
            ]
            ch *= 2
        layers += [ResidualBlock(ch) for _ in range(num_residual_blocks)]
        for _ in range(2):  # upsample
            layers += [
                nn.Upsample(scale_factor=2),
                nn.Conv2d(ch, ch // 2, 3, 1, 1),
                nn.InstanceNorm2d(ch // 2),
                nn.ReLU(inplace=True),
            ]
            ch //= 2
        layers += [nn.ReflectionPad2d(3), nn.Conv2d(ch, channels, 7), nn.Tanh()]
        self.model = nn.Sequential(*layers)

    def forward(self, x):
        return self.model(x)


class Discriminator(nn.Module):
    def __init__(self, channels=3, base=64):
        super().__init__()

        def block(cin, cout, norm=True):
            layers = [nn.Conv2d(cin, cout, 4, 2, 1)]
            if norm:
                layers.append(nn.InstanceNorm2d(cout))
            layers.append(nn.LeakyReLU(0.2, inplace=True))
            return layers

        self.model = nn.Sequential(
            *block(channels, base, norm=False),
            *block(base, base * 2),
            *block(base * 2, base * 4),
            *block(base * 4, base * 8),
            nn.ZeroPad2d((1, 0, 1, 0)),
            nn.Conv2d(base * 8, 1, 4, padding=1),
        )

    def forward(self, x):
        return self.model(x)
