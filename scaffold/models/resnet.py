"""In-repo ResNet-50 (no torchvision in this environment).

The headline-benchmark model (BASELINE.json: samples/sec, ResNet-50
bf16 DDP).  Convolutions run through torch -> MIOpen (the library conv
path; SURVEY §2.3 K5 note); the loss is the fused softmax-cross-entropy
op and the optimizer step is the fused AdamW HIP kernel.

Standard v1.5 architecture: stride-2 in the 3x3 of each downsampling
bottleneck; BN layers kept in fp32 internally by torch's own BN.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


class Fp32BatchNorm(nn.BatchNorm2d):
    """Fused BatchNorm (+optional ReLU, +optional residual add).

    fp32 parameters/stats regardless of activation dtype (Trainer's
    _to_mixed keeps BatchNorm2d subclasses fp32).  On GPU in training
    mode this runs the fused bf16-NHWC HIP kernels (stats + apply in 3
    activation passes, ReLU and the block's residual add folded in) —
    MIOpen's fp32-only BN plus its cast copies plus the separate
    add/relu passes were ~57% of the r1 headline step
    (profiles/resnet_kstats_r2.txt).  CPU and eval mode use the fp32
    eager reference."""

    def __init__(self, ch: int, relu: bool = False):
        super().__init__(ch)
        self.relu = relu

    def forward(self, x, res=None):
        if self.training and self.num_batches_tracked is not None:
            self.num_batches_tracked += 1
        return ops.batch_norm_act(
            x, res, self.weight, self.bias, self.running_mean,
            self.running_var, self.momentum, self.eps, self.relu,
            self.training)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, width: int, stride: int = 1, downsample: nn.Module = None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = Fp32BatchNorm(width, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1, bias=False)
        self.bn2 = Fp32BatchNorm(width, relu=True)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = Fp32BatchNorm(out_ch, relu=True)  # fused add+relu
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        # bn3 folds the residual add AND the final relu into its apply
        return self.bn3(self.conv3(out), res=identity)


class ResNet50(nn.Module):
    layers = (3, 4, 6, 3)

    def __init__(self, cfg):
        super().__init__()
        num_classes = cfg.model.num_classes
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = Fp32BatchNorm(64, relu=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, self.layers[0], stride=1)
        self.layer2 = self._make_layer(128, self.layers[1], stride=2)
        self.layer3 = self._make_layer(256, self.layers[2], stride=2)
        self.layer4 = self._make_layer(512, self.layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        # zero-init last BN of each block (standard recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, width: int, blocks: int, stride: int) -> nn.Sequential:
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                Fp32BatchNorm(out_ch),
            )
        layers = [Bottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(out_ch, width))
        return nn.Sequential(*layers)

    def features(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return torch.flatten(self.avgpool(x), 1)

    def forward(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        logits = self.fc(self.features(x))
        return ops.softmax_cross_entropy(logits, y)
