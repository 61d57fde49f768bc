"""The GeoMX example CNN (examples/cnn.py:56-63), generalised to the
benchmark input shape.

Reference architecture: Conv(16,k5,relu) - MaxPool(2,2) -
Conv(32,k5,relu) - MaxPool(2,2) - Dense(256,relu) - Dense(128,relu) -
Dense(num_classes). The reference trains it on 1x28x28 MNIST; the
project benchmark (BASELINE.json) runs the same topology on synthetic
3x224x224, which makes the first Dense a wide VGG-style classifier
head.
"""

from __future__ import annotations

import torch
import torch.nn as nn


class GeoCNN(nn.Module):
    def __init__(self, in_channels: int = 3, image_size: int = 224,
                 num_classes: int = 10):
        super().__init__()
        from ..ops.conv import GeoConv5Pool
        self.features = nn.Sequential(
            # both conv stages: conv+bias+relu+maxpool in ONE gfx950
            # kernel each (no full-resolution conv output in HBM)
            GeoConv5Pool(in_channels, 16),
            GeoConv5Pool(16, 32),
        )
        with torch.no_grad():
            probe = torch.zeros(1, in_channels, image_size, image_size)
            flat = self.features(probe).numel()
        self.classifier = nn.Sequential(
            nn.Flatten(),
            nn.Linear(flat, 256),
            nn.ReLU(inplace=True),
            nn.Linear(256, 128),
            nn.ReLU(inplace=True),
            nn.Linear(128, num_classes),
        )

    def forward(self, x):
        return self.classifier(self.features(x))


def geo_cnn(in_channels: int = 3, image_size: int = 224,
            num_classes: int = 10) -> GeoCNN:
    return GeoCNN(in_channels, image_size, num_classes)
