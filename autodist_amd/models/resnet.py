"""ResNet family (v1.5, bottleneck) — the headline benchmark model.

Reference workloads: examples/benchmark/imagenet.py benchmarks ResNet101 /
VGG16 / DenseNet121 / InceptionV3; BASELINE.json's headline metric is
images/sec ResNet-50 AllReduce bf16. Implemented from the architecture
definition (He et al. 2015; v1.5 stride-in-3x3 variant) — torchvision is not
in the image.

MI355X notes: intended to run with channels_last memory format + bf16
autocast (MIOpen picks NHWC-friendly algorithms), parameters fp32 and
flat-bucketed by the engine.
"""
import torch
import torch.nn as nn


def _make_bn(ch, fused, relu):
    if fused:
        from autodist_amd.ops.fused_bn import FusedBatchNorm2d
        return FusedBatchNorm2d(ch, relu=relu)
    return nn.BatchNorm2d(ch)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1, downsample=None, fused=False):
        super().__init__()
        out_ch = width * self.expansion
        self.fused = fused
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = _make_bn(width, fused, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = _make_bn(width, fused, relu=True)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = _make_bn(out_ch, fused, relu=True)  # fused: bn+add+relu
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        if self.fused:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            z = self.conv3(out)
            identity = x if self.downsample is None else self.downsample(x)
            return self.bn3.forward_add(z, identity)
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_ch, width, stride=1, downsample=None, fused=False):
        super().__init__()
        self.fused = fused
        self.conv1 = nn.Conv2d(in_ch, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn1 = _make_bn(width, fused, relu=True)
        self.conv2 = nn.Conv2d(width, width, 3, padding=1, bias=False)
        self.bn2 = _make_bn(width, fused, relu=True)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        if self.fused:
            out = self.bn1(self.conv1(x))
            z = self.conv2(out)
            identity = x if self.downsample is None else self.downsample(x)
            return self.bn2.forward_add(z, identity)
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, fused=False):
        super().__init__()
        self.in_ch = 64
        self.fused = fused
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = _make_bn(64, fused, relu=True)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        from autodist_amd.ops.fused_bn import FusedBatchNorm2d
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, (nn.BatchNorm2d, FusedBatchNorm2d)):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, block, width, blocks, stride=1):
        downsample = None
        out_ch = width * block.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                _make_bn(out_ch, self.fused, relu=False))
        layers = [block(self.in_ch, width, stride, downsample,
                        fused=self.fused)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(block(self.in_ch, width, fused=self.fused))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.bn1(self.conv1(x))
        if not self.fused:
            x = self.relu(x)
        x = self.maxpool(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18(num_classes=1000, fused=False):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, fused=fused)


def resnet50(num_classes=1000, fused=False):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, fused=fused)


def resnet101(num_classes=1000, fused=False):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, fused=fused)
