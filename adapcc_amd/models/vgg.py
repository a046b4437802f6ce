"""VGG-16 (reference: train_ddp.py's canonical DDP workload used
torchvision VGG16; self-contained here — no torchvision in the image)."""

from __future__ import annotations

import torch
import torch.nn as nn

_CFG16 = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class VGG16(nn.Module):
    def __init__(self, num_classes: int = 1000, in_size: int = 224):
        super().__init__()
        layers = []
        c = 3
        for v in _CFG16:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(c, v, 3, padding=1), nn.ReLU(inplace=True)]
                c = v
        self.features = nn.Sequential(*layers)
        spatial = in_size // 32
        self.classifier = nn.Sequential(
            nn.Linear(512 * spatial * spatial, 4096), nn.ReLU(inplace=True),
            nn.Dropout(0.5),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(0.5),
            nn.Linear(4096, num_classes),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.features(x)
        return self.classifier(torch.flatten(x, 1))
