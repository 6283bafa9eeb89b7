"""ResNet-18 image predictor (torchvision is not in the image, so the
standard architecture is defined here directly) and the superpixel masking
config for KernelSHAP on images — BASELINE.json config 5: "ResNet-18 image
predictor on 224x224 superpixel masks (large perturbation-batch predict
path)".

An image instance is the flattened (3*H*W) pixel vector; explained features
are PxP superpixel patches, i.e. groups of pixel columns, so the generic
masked-background synthesis kernel (K3') and the torch-predictor path apply
unchanged: masked-out patches are replaced by the background image's pixels.
"""
from __future__ import annotations

from typing import List, Tuple

import numpy as np

__all__ = ["resnet18", "make_superpixel_problem"]


def resnet18(num_classes: int = 10, seed: int = 0):
    import torch
    from torch import nn

    class BasicBlock(nn.Module):
        def __init__(self, cin, cout, stride=1):
            super().__init__()
            self.conv1 = nn.Conv2d(cin, cout, 3, stride, 1, bias=False)
            self.bn1 = nn.BatchNorm2d(cout)
            self.conv2 = nn.Conv2d(cout, cout, 3, 1, 1, bias=False)
            self.bn2 = nn.BatchNorm2d(cout)
            self.relu = nn.ReLU(inplace=True)
            self.down = None
            if stride != 1 or cin != cout:
                self.down = nn.Sequential(
                    nn.Conv2d(cin, cout, 1, stride, bias=False),
                    nn.BatchNorm2d(cout),
                )

        def forward(self, x):
            idt = x if self.down is None else self.down(x)
            out = self.relu(self.bn1(self.conv1(x)))
            out = self.bn2(self.conv2(out))
            return self.relu(out + idt)

    class ResNet18(nn.Module):
        def __init__(self):
            super().__init__()
            self.stem = nn.Sequential(
                nn.Conv2d(3, 64, 7, 2, 3, bias=False),
                nn.BatchNorm2d(64),
                nn.ReLU(inplace=True),
                nn.MaxPool2d(3, 2, 1),
            )
            layers = []
            cin = 64
            for cout, stride in [(64, 1), (64, 1), (128, 2), (128, 1),
                                 (256, 2), (256, 1), (512, 2), (512, 1)]:
                layers.append(BasicBlock(cin, cout, stride))
                cin = cout
            self.layers = nn.Sequential(*layers)
            self.pool = nn.AdaptiveAvgPool2d(1)
            self.fc = nn.Linear(512, num_classes)

        def forward(self, x):
            # accepts flattened (n, 3*H*W) rows from the synth kernel
            if x.dim() == 2:
                hw = int((x.shape[1] // 3) ** 0.5)
                x = x.view(-1, 3, hw, hw)
            if self.stem[0].weight.is_contiguous(
                memory_format=torch.channels_last
            ):
                # match the NHWC weight layout once at the input instead of
                # per-conv internal transposes
                x = x.contiguous(memory_format=torch.channels_last)
            x = self.stem(x)
            x = self.layers(x)
            x = self.pool(x).flatten(1)
            return torch.softmax(self.fc(x), dim=-1)

    torch.manual_seed(seed)
    return ResNet18().eval()


def make_superpixel_problem(
    n_instances: int = 4,
    hw: int = 224,
    patch: int = 32,
    seed: int = 0,
) -> Tuple[np.ndarray, np.ndarray, List[List[int]], List[str]]:
    """Synthetic image instances + blurred-background + superpixel groups.

    Returns (X (B, 3*hw*hw), background (1, 3*hw*hw), groups, group_names).
    Groups are (hw/patch)^2 patches; the background is a box-blurred version
    of the mean image (the standard image-masking baseline).
    """
    rng = np.random.Generator(np.random.Philox(key=[seed, 0x1CE]))
    imgs = rng.random((n_instances, 3, hw, hw), dtype=np.float32)
    mean = imgs.mean(axis=0)
    # cheap box blur as background
    k = 8
    pad = np.pad(mean, ((0, 0), (k, k), (k, k)), mode="edge")
    blur = np.zeros_like(mean)
    for dy in (-k, 0, k):
        for dx in (-k, 0, k):
            blur += pad[:, k + dy : k + dy + hw, k + dx : k + dx + hw]
    blur /= 9.0
    g = hw // patch
    groups: List[List[int]] = []
    names: List[str] = []
    idx = np.arange(3 * hw * hw).reshape(3, hw, hw)
    for py in range(g):
        for px in range(g):
            cols = idx[:, py * patch : (py + 1) * patch, px * patch : (px + 1) * patch]
            groups.append(cols.ravel().tolist())
            names.append(f"patch_{py}_{px}")
    return (
        imgs.reshape(n_instances, -1).astype(np.float64),
        blur.reshape(1, -1).astype(np.float64),
        groups,
        names,
    )
