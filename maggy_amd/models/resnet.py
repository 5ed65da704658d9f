"""ResNet v1.5 (bottleneck) — the benchmark flagship model.

Self-contained implementation (torchvision is not a dependency of this
framework).  ResNet-50/101/152 with the standard v1.5 stride placement
(stride 2 in the 3x3 of the bottleneck).  Used by BASELINE.json configs 2/3
(ResNet-50 synthetic 224x224 HPO/ASHA) and the samples/sec benchmark.

MI355X notes: run under channels_last + bf16 autocast so MIOpen picks its
NHWC kernels; the model is a plain nn.Module — the fused optimizer and DDP
comm hooks live in maggy_amd.ops / maggy_amd.parallel.
"""
import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1, downsample=None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes=1000, in_ch=3):
        super().__init__()
        self.in_planes = 64
        self.conv1 = nn.Conv2d(in_ch, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
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
                nn.BatchNorm2d(out_ch),
            )
        layers = [Bottleneck(self.in_planes, width, stride, downsample)]
        self.in_planes = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_planes, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
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
