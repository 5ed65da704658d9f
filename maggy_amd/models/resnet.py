"""ResNet v1.5 (bottleneck) — the benchmark flagship model.

Self-contained implementation (torchvision is not a dependency of this
framework).  ResNet-50/101/152 with the standard v1.5 stride placement
(stride 2 in the 3x3 of the bottleneck).  Used by BASELINE.json configs 2/3
(ResNet-50 synthetic 224x224 HPO/ASHA) and the samples/sec benchmark.

MI355X design: every ``BN -> ReLU`` and ``BN -> add -> ReLU`` pattern is a
single MaggyBatchNorm2d module (ops/fused_bn.py) so the hand-written fused
HIP kernels run the whole normalization/activation/residual glue in bf16
channels_last (the eager path was 53% of the step, profiles/r01).  On CPU
or without the extension the module falls back to torch batch_norm, so the
architecture and parameter count are unchanged (matches torchvision's
25,557,032 params for ResNet-50).
"""
import torch
import torch.nn as nn

from maggy_amd.ops.fused_bn import MaggyBatchNorm2d


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1, downsample=None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = MaggyBatchNorm2d(width, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = MaggyBatchNorm2d(width, relu=True)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        # bn3 fuses the residual add + final relu
        self.bn3 = MaggyBatchNorm2d(out_ch, relu=True)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        return self.bn3(self.conv3(out), residual=identity)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes=1000, in_ch=3):
        super().__init__()
        self.in_planes = 64
        self.conv1 = nn.Conv2d(in_ch, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = MaggyBatchNorm2d(64, relu=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)
        # zero-init the last BN of each block (standard recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, width, blocks, stride=1):
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_planes != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_planes, out_ch, 1, stride=stride,
                          bias=False),
                MaggyBatchNorm2d(out_ch, relu=False),
            )
        layers = [Bottleneck(self.in_planes, width, stride, downsample)]
        self.in_planes = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_planes, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        return self.fc(torch.flatten(x, 1))


def resnet50(num_classes=1000):
    return ResNet([3, 4, 6, 3], num_classes=num_classes)


def resnet101(num_classes=1000):
    return ResNet([3, 4, 23, 3], num_classes=num_classes)


def resnet152(num_classes=1000):
    return ResNet([3, 8, 36, 3], num_classes=num_classes)


def resnet18_thin(num_classes=10):
    """Tiny bottleneck ResNet for CPU tests."""
    return ResNet([1, 1, 1, 1], num_classes=num_classes)
