"""DenseNet (Huang et al. 2016) — reference benchmark model
(examples/benchmark/imagenet.py). BN-ReLU-Conv pre-activation ordering maps
directly onto the fused gfx950 BN+ReLU kernels (`fused=True`)."""
import torch
import torch.nn as nn


class _BNReLU(nn.Module):
    """BN+ReLU with state_dict keys identical across fused/unfused variants
    (key: '<name>.bn.*')."""

    def __init__(self, ch, fused):
        super().__init__()
        self.fused = fused
        if fused:
            from autodist_amd.ops.fused_bn import FusedBatchNorm2d
            self.bn = FusedBatchNorm2d(ch, relu=True)
        else:
            self.bn = nn.BatchNorm2d(ch)

    def forward(self, x):
        x = self.bn(x)
        return x if self.fused else torch.relu(x)


def _bn_relu(ch, fused):
    return _BNReLU(ch, fused)


class DenseLayer(nn.Module):
    def __init__(self, in_ch, growth, bn_size=4, fused=False):
        super().__init__()
        self.norm1 = _bn_relu(in_ch, fused)
        self.conv1 = nn.Conv2d(in_ch, bn_size * growth, 1, bias=False)
        self.norm2 = _bn_relu(bn_size * growth, fused)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        out = self.conv1(self.norm1(x))
        return self.conv2(self.norm2(out))


class DenseBlock(nn.Module):
    def __init__(self, num_layers, in_ch, growth, bn_size=4, fused=False):
        super().__init__()
        self.layers = nn.ModuleList(
            DenseLayer(in_ch + i * growth, growth, bn_size, fused)
            for i in range(num_layers))

    def forward(self, x):
        features = [x]
        for layer in self.layers:
            features.append(layer(torch.cat(features, 1)))
        return torch.cat(features, 1)


class Transition(nn.Module):
    def __init__(self, in_ch, out_ch, fused=False):
        super().__init__()
        self.norm = _bn_relu(in_ch, fused)
        self.conv = nn.Conv2d(in_ch, out_ch, 1, bias=False)
        self.pool = nn.AvgPool2d(2, 2)

    def forward(self, x):
        return self.pool(self.conv(self.norm(x)))


class DenseNet(nn.Module):
    def __init__(self, growth=32, block_config=(6, 12, 24, 16),
                 init_ch=64, bn_size=4, num_classes=1000, fused=False):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(3, init_ch, 7, stride=2, padding=3, bias=False))
        self.stem_norm = _bn_relu(init_ch, fused)
        self.stem_pool = nn.MaxPool2d(3, stride=2, padding=1)
        ch = init_ch
        blocks = []
        for i, n in enumerate(block_config):
            blocks.append(DenseBlock(n, ch, growth, bn_size, fused))
            ch += n * growth
            if i != len(block_config) - 1:
                blocks.append(Transition(ch, ch // 2, fused))
                ch //= 2
        self.blocks = nn.Sequential(*blocks)
        self.final_norm = _bn_relu(ch, fused)
        self.classifier = nn.Linear(ch, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight)
            elif isinstance(m, nn.Linear):
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.stem_pool(self.stem_norm(self.stem(x)))
        x = self.final_norm(self.blocks(x))
        x = torch.nn.functional.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.classifier(x)


def densenet121(num_classes=1000, fused=False):
    return DenseNet(32, (6, 12, 24, 16), 64, num_classes=num_classes,
                    fused=fused)
