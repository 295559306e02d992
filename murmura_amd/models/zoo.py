"""Model zoo: MLPs/CNNs matching the reference's families and size ladder
(reference: murmura/examples/leaf/models.py:12-216, datasets.py:204-297), plus
the ResNet-18-sized benchmark CNN named by BASELINE.json config 2.

All models are plain float modules; binding to flat buffers happens in
FlatParamStore, dtype (fp32/bf16) is chosen by the node.
"""

from __future__ import annotations

from typing import Callable, Dict

import torch
from torch import Tensor, nn

from murmura_amd.ops.fused_bn import (
    MurmuraBatchNorm2d,
    MurmuraBNAddReLU,
    MurmuraBNReLU,
)
from murmura_amd.ops.fused_conv import MurmuraConv3x3


class SimpleMLP(nn.Module):
    """Tiny MLP for synthetic-data tests (the reference's programmatic example
    uses an equivalent, examples/simple_programmatic.py:43-96)."""

    def __init__(self, in_features: int = 20, hidden: int = 32, num_classes: int = 4):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(in_features, hidden),
            nn.ReLU(),
            nn.Linear(hidden, num_classes),
        )

    def forward(self, x: Tensor) -> Tensor:
        return self.net(x)


class FEMNISTModel(nn.Module):
    """LEAF FEMNIST CNN: conv5x5(32)-pool-conv5x5(64)-pool-fc(H)-fc(62).

    hidden=2048 gives the ~6.5M-param baseline (reference:
    examples/leaf/datasets.py:204-232); the tiny/small/large/xlarge variants
    scale the fc width to ~0.2M / 0.8M / 13M / 26M params
    (examples/leaf/models.py:12-216).
    """

    def __init__(self, hidden: int = 2048, num_classes: int = 62):
        super().__init__()
        self.conv1 = nn.Conv2d(1, 32, kernel_size=5, padding=2)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=5, padding=2)
        self.pool = nn.MaxPool2d(2)
        self.fc1 = nn.Linear(7 * 7 * 64, hidden)
        self.fc2 = nn.Linear(hidden, num_classes)

    def forward(self, x: Tensor) -> Tensor:
        x = self.pool(torch.relu(self.conv1(x)))
        x = self.pool(torch.relu(self.conv2(x)))
        x = x.flatten(1)
        x = torch.relu(self.fc1(x))
        return self.fc2(x)


_FEMNIST_VARIANTS: Dict[str, int] = {
    "tiny": 64,
    "small": 256,
    "baseline": 2048,
    "large": 4096,
    "xlarge": 8192,
}


def get_model_variant(variant: str = "baseline", num_classes: int = 62) -> FEMNISTModel:
    if variant not in _FEMNIST_VARIANTS:
        raise ValueError(
            f"unknown FEMNIST variant {variant!r}; options: {sorted(_FEMNIST_VARIANTS)}"
        )
    return FEMNISTModel(hidden=_FEMNIST_VARIANTS[variant], num_classes=num_classes)


class CelebAModel(nn.Module):
    """LeNet-style CNN for 84x84 RGB CelebA binary attribute classification
    (reference: examples/leaf/datasets.py:235-297)."""

    def __init__(self, num_classes: int = 2):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(3, 32, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),  # 42
            nn.Conv2d(32, 32, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),  # 21
            nn.Conv2d(32, 32, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),  # 10
            nn.Conv2d(32, 32, 3, padding=1),
            nn.ReLU(),
            nn.MaxPool2d(2),  # 5
        )
        self.classifier = nn.Linear(32 * 5 * 5, num_classes)

    def forward(self, x: Tensor) -> Tensor:
        return self.classifier(self.features(x).flatten(1))


# ----------------------------------------------------------------- ResNet-18
class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__()
        # stride-1 3x3 convs route their weight gradient through the K15
        # MFMA kernel on the shapes where it measures faster (ops/fused_conv)
        self.conv1 = (MurmuraConv3x3(in_ch, out_ch, stride=1) if stride == 1
                      else nn.Conv2d(in_ch, out_ch, 3, stride=stride, padding=1,
                                     bias=False))
        self.bn1 = MurmuraBNReLU(out_ch)  # ReLU folded into the BN kernels
        self.conv2 = MurmuraConv3x3(out_ch, out_ch, stride=1)
        # block tail relu(bn2(conv2) + shortcut) fused into one kernel each
        # direction (residual grad comes from the same backward launch)
        self.bn2 = MurmuraBNAddReLU(out_ch)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_ch != out_ch:
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                MurmuraBatchNorm2d(out_ch),
            )

    def forward(self, x: Tensor) -> Tensor:
        out = self.bn1(self.conv1(x))  # BN+ReLU fused
        return self.bn2(self.conv2(out), res=self.shortcut(x))


class ResNet18(nn.Module):
    """ResNet-18 with a CIFAR-style 3x3 stem (~11.2M params) — the
    "ResNet-18-sized CNN" of BASELINE.json config 2."""

    def __init__(self, num_classes: int = 10, in_channels: int = 3):
        super().__init__()
        self.conv1 = nn.Conv2d(in_channels, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = MurmuraBNReLU(64)  # ReLU folded into the BN kernels
        layers = []
        in_ch = 64
        for out_ch, stride in [(64, 1), (64, 1), (128, 2), (128, 1),
                               (256, 2), (256, 1), (512, 2), (512, 1)]:
            layers.append(BasicBlock(in_ch, out_ch, stride))
            in_ch = out_ch
        self.layers = nn.Sequential(*layers)
        self.fc = nn.Linear(512, num_classes)

    def forward(self, x: Tensor) -> Tensor:
        out = self.bn1(self.conv1(x))  # BN+ReLU fused
        out = self.layers(out)
        out = torch.nn.functional.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


class WideMLP(nn.Module):
    """~100M-param MLP (in 4096 -> 12288 -> 4096 -> classes): the
    bandwidth-bound Sketchguard showcase model (BASELINE.json config 4) —
    P ~ 100M floats means each neighbor state is 400 MB fp32 / 200 MB bf16
    on the xGMI wire."""

    def __init__(self, in_features: int = 4096, hidden: int = 12288,
                 num_classes: int = 62):
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(in_features, hidden),
            nn.ReLU(),
            nn.Linear(hidden, in_features),
            nn.ReLU(),
            nn.Linear(in_features, num_classes),
        )

    def forward(self, x: Tensor) -> Tensor:
        return self.net(x)


def count_params(model: nn.Module) -> int:
    return sum(p.numel() for p in model.parameters())


MODEL_FACTORIES: Dict[str, Callable[..., nn.Module]] = {
    "models.mlp": SimpleMLP,
    "models.femnist": FEMNISTModel,
    "models.celeba": CelebAModel,
    "models.resnet18": ResNet18,
    "models.widemlp": WideMLP,
}
