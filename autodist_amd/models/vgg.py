"""VGG (Simonyan & Zisserman 2014) — reference benchmark model
(examples/benchmark/imagenet.py). Classic config D (VGG16); optional BN
variant uses the fused gfx950 BN+ReLU kernels."""
import torch.nn as nn

_CFG_D = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class VGG(nn.Module):
    def __init__(self, cfg=None, num_classes=1000, batch_norm=False,
                 fused=False):
        super().__init__()
        cfg = cfg or _CFG_D
        layers = []
        in_ch = 3
        for v in cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
                continue
            layers.append(nn.Conv2d(in_ch, v, 3, padding=1,
                                    bias=not batch_norm))
            if batch_norm:
                if fused:
                    from autodist_amd.ops.fused_bn import FusedBatchNorm2d
                    layers.append(FusedBatchNorm2d(v, relu=True))
                else:
                    layers += [nn.BatchNorm2d(v), nn.ReLU(inplace=True)]
            else:
                layers.append(nn.ReLU(inplace=True))
            in_ch = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(inplace=True),
            nn.Dropout(0.5),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(0.5),
            nn.Linear(4096, num_classes))
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
            elif isinstance(m, nn.Linear):
                nn.init.normal_(m.weight, 0, 0.01)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = self.avgpool(self.features(x)).flatten(1)
        return self.classifier(x)


def vgg16(num_classes=1000, batch_norm=False, fused=False):
    return VGG(num_classes=num_classes, batch_norm=batch_norm, fused=fused)
