"""ResNet-50 (v1.5, stride-2 in the 3x3 of the bottleneck) — the headline
benchmark model (BASELINE.json metric: images/sec ResNet-50 synthetic).

Own implementation (torchvision is not in this image); standard
architecture: He et al., Deep Residual Learning for Image Recognition.
"""
import torch
import torch.nn as nn

from axon.ops import FusedBatchNorm2d


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1, downsample=None,
                 norm_layer=FusedBatchNorm2d):
        super().__init__()
        out_ch = width * self.expansion
        fuses = norm_layer is FusedBatchNorm2d
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = (norm_layer(width, fuse_relu=True) if fuses
                    else norm_layer(width))
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = (norm_layer(width, fuse_relu=True) if fuses
                    else norm_layer(width))
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = norm_layer(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self._bn_fuses_relu = fuses
        self.downsample = downsample

    def forward(self, x):
        identity = x
        # bn1/bn2 fold their trailing ReLU into the fused NHWC BN kernel;
        # bn3 additionally folds the residual add + final ReLU
        # (axon/ops/batchnorm.py forward_add_relu) — the whole bottleneck
        # tail is one pass over the activations.
        out = self.bn1(self.conv1(x))
        if not self._bn_fuses_relu:
            out = self.relu(out)
        out = self.bn2(self.conv2(out))
        if not self._bn_fuses_relu:
            out = self.relu(out)
        out = self.conv3(out)
        if self.downsample is not None:
            identity = self.downsample(x)
        if self._bn_fuses_relu:
            return self.bn3.forward_add_relu(out, identity)
        return self.relu(self.bn3(out) + identity)


class ResNet(nn.Module):
    def __init__(self, layers, num_classes=1000, norm_layer=FusedBatchNorm2d):
        super().__init__()
        self._norm = norm_layer
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = norm_layer(64, fuse_relu=True)             if norm_layer is FusedBatchNorm2d else norm_layer(64)
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
        # zero-init last BN gamma in each block (standard trick)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)

    def _make_layer(self, width, blocks, stride=1):
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                self._norm(out_ch),
            )
        layers = [Bottleneck(self.in_ch, width, stride, downsample,
                             norm_layer=self._norm)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_ch, width, norm_layer=self._norm))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.bn1(self.conv1(x))
        if not getattr(self.bn1, "fuse_relu", False):
            x = self.relu(x)
        x = self.maxpool(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes=1000, **kw):
    return ResNet([3, 4, 6, 3], num_classes=num_classes, **kw)


def resnet101(num_classes=1000, **kw):
    return ResNet([3, 4, 23, 3], num_classes=num_classes, **kw)
