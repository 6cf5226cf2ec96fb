"""Inception v3 (Szegedy et al. 2015) — reference benchmark model
(examples/benchmark/imagenet.py). Conv-BN-ReLU cells use the fused gfx950
BN kernels (`fused=True`); aux head omitted (benchmark parity: throughput of
the main tower)."""
import torch
import torch.nn as nn


class ConvBN(nn.Module):
    def __init__(self, in_ch, out_ch, fused=False, **conv_kw):
        super().__init__()
        self.conv = nn.Conv2d(in_ch, out_ch, bias=False, **conv_kw)
        if fused:
            from autodist_amd.ops.fused_bn import FusedBatchNorm2d
            self.bn = FusedBatchNorm2d(out_ch, relu=True)
            self._act = None
        else:
            self.bn = nn.BatchNorm2d(out_ch, eps=1e-3)
            self._act = nn.ReLU(inplace=True)

    def forward(self, x):
        x = self.bn(self.conv(x))
        return self._act(x) if self._act is not None else x


class InceptionA(nn.Module):
    def __init__(self, in_ch, pool_ch, fused=False):
        super().__init__()
        f = fused
        self.b1 = ConvBN(in_ch, 64, f, kernel_size=1)
        self.b5 = nn.Sequential(ConvBN(in_ch, 48, f, kernel_size=1),
                                ConvBN(48, 64, f, kernel_size=5, padding=2))
        self.b3 = nn.Sequential(ConvBN(in_ch, 64, f, kernel_size=1),
                                ConvBN(64, 96, f, kernel_size=3, padding=1),
                                ConvBN(96, 96, f, kernel_size=3, padding=1))
        self.pool = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                  ConvBN(in_ch, pool_ch, f, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b5(x), self.b3(x), self.pool(x)], 1)


class InceptionB(nn.Module):
    def __init__(self, in_ch, fused=False):
        super().__init__()
        f = fused
        self.b3 = ConvBN(in_ch, 384, f, kernel_size=3, stride=2)
        self.b3d = nn.Sequential(ConvBN(in_ch, 64, f, kernel_size=1),
                                 ConvBN(64, 96, f, kernel_size=3, padding=1),
                                 ConvBN(96, 96, f, kernel_size=3, stride=2))
        self.pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b3(x), self.b3d(x), self.pool(x)], 1)


class InceptionC(nn.Module):
    def __init__(self, in_ch, c7, fused=False):
        super().__init__()
        f = fused
        self.b1 = ConvBN(in_ch, 192, f, kernel_size=1)
        self.b7 = nn.Sequential(
            ConvBN(in_ch, c7, f, kernel_size=1),
            ConvBN(c7, c7, f, kernel_size=(1, 7), padding=(0, 3)),
            ConvBN(c7, 192, f, kernel_size=(7, 1), padding=(3, 0)))
        self.b7d = nn.Sequential(
            ConvBN(in_ch, c7, f, kernel_size=1),
            ConvBN(c7, c7, f, kernel_size=(7, 1), padding=(3, 0)),
            ConvBN(c7, c7, f, kernel_size=(1, 7), padding=(0, 3)),
            ConvBN(c7, c7, f, kernel_size=(7, 1), padding=(3, 0)),
            ConvBN(c7, 192, f, kernel_size=(1, 7), padding=(0, 3)))
        self.pool = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                  ConvBN(in_ch, 192, f, kernel_size=1))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b7(x), self.b7d(x),
                          self.pool(x)], 1)


class InceptionD(nn.Module):
    def __init__(self, in_ch, fused=False):
        super().__init__()
        f = fused
        self.b3 = nn.Sequential(ConvBN(in_ch, 192, f, kernel_size=1),
                                ConvBN(192, 320, f, kernel_size=3, stride=2))
        self.b7 = nn.Sequential(
            ConvBN(in_ch, 192, f, kernel_size=1),
            ConvBN(192, 192, f, kernel_size=(1, 7), padding=(0, 3)),
            ConvBN(192, 192, f, kernel_size=(7, 1), padding=(3, 0)),
            ConvBN(192, 192, f, kernel_size=3, stride=2))
        self.pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b3(x), self.b7(x), self.pool(x)], 1)


class InceptionE(nn.Module):
    def __init__(self, in_ch, fused=False):
        super().__init__()
        f = fused
        self.b1 = ConvBN(in_ch, 320, f, kernel_size=1)
        self.b3_stem = ConvBN(in_ch, 384, f, kernel_size=1)
        self.b3_a = ConvBN(384, 384, f, kernel_size=(1, 3), padding=(0, 1))
        self.b3_b = ConvBN(384, 384, f, kernel_size=(3, 1), padding=(1, 0))
        self.b3d_stem = nn.Sequential(
            ConvBN(in_ch, 448, f, kernel_size=1),
            ConvBN(448, 384, f, kernel_size=3, padding=1))
        self.b3d_a = ConvBN(384, 384, f, kernel_size=(1, 3), padding=(0, 1))
        self.b3d_b = ConvBN(384, 384, f, kernel_size=(3, 1), padding=(1, 0))
        self.pool = nn.Sequential(nn.AvgPool2d(3, 1, 1),
                                  ConvBN(in_ch, 192, f, kernel_size=1))

    def forward(self, x):
        s = self.b3_stem(x)
        d = self.b3d_stem(x)
        return torch.cat([self.b1(x), self.b3_a(s), self.b3_b(s),
                          self.b3d_a(d), self.b3d_b(d), self.pool(x)], 1)


class InceptionV3(nn.Module):
    def __init__(self, num_classes=1000, fused=False, dropout=0.5):
        super().__init__()
        f = fused
        self.stem = nn.Sequential(
            ConvBN(3, 32, f, kernel_size=3, stride=2),
            ConvBN(32, 32, f, kernel_size=3),
            ConvBN(32, 64, f, kernel_size=3, padding=1),
            nn.MaxPool2d(3, 2),
            ConvBN(64, 80, f, kernel_size=1),
            ConvBN(80, 192, f, kernel_size=3),
            nn.MaxPool2d(3, 2))
        self.mixed = nn.Sequential(
            InceptionA(192, 32, f), InceptionA(256, 64, f),
            InceptionA(288, 64, f),
            InceptionB(288, f),
            InceptionC(768, 128, f), InceptionC(768, 160, f),
            InceptionC(768, 160, f), InceptionC(768, 192, f),
            InceptionD(768, f),
            InceptionE(1280, f), InceptionE(2048, f))
        self.dropout = nn.Dropout(dropout)
        self.fc = nn.Linear(2048, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight)

    def forward(self, x):
        x = self.mixed(self.stem(x))
        x = torch.nn.functional.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(self.dropout(x))


def inception_v3(num_classes=1000, fused=False):
    return InceptionV3(num_classes=num_classes, fused=fused)
