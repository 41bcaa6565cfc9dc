"""DenseNet-121/169/201 (reference matrix row: densenet201 bs32).

``fused_bn=True`` uses the CDNA4 fused NHWC BN+ReLU kernel for the
BN->ReLU->Conv pattern (~200 sites in densenet201); identical numerics and
state-dict keys (ops/fused_bn.py)."""
import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["densenet121", "densenet169", "densenet201"]


def _bn_relu(c, fused):
    if fused:
        from ..ops.fused_bn import FusedBNAct2d
        return FusedBNAct2d(c, relu=True)
    return None  # caller composes nn.BatchNorm2d + F.relu


class DenseLayer(nn.Module):
    def __init__(self, cin, growth, bn_size=4, fused_bn=False):
        super().__init__()
        self.fused = fused_bn
        self.norm1 = _bn_relu(cin, fused_bn) or nn.BatchNorm2d(cin)
        self.conv1 = nn.Conv2d(cin, bn_size * growth, 1, bias=False)
        self.norm2 = _bn_relu(bn_size * growth, fused_bn) or             nn.BatchNorm2d(bn_size * growth)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        if self.fused:
            out = self.conv1(self.norm1(x))
            out = self.conv2(self.norm2(out))
        else:
            out = self.conv1(F.relu(self.norm1(x)))
            out = self.conv2(F.relu(self.norm2(out)))
        return torch.cat([x, out], 1)


class Transition(nn.Module):
    def __init__(self, cin, cout, fused_bn=False):
        super().__init__()
        self.fused = fused_bn
        self.norm = _bn_relu(cin, fused_bn) or nn.BatchNorm2d(cin)
        self.conv = nn.Conv2d(cin, cout, 1, bias=False)

    def forward(self, x):
        h = self.norm(x) if self.fused else F.relu(self.norm(x))
        return F.avg_pool2d(self.conv(h), 2)


class DenseNet(nn.Module):
    def __init__(self, block_cfg, growth=32, init_ch=64, num_classes=1000,
                 fused_bn=False):
        super().__init__()
        stem_bn = _bn_relu(init_ch, fused_bn)
        layers = [nn.Conv2d(3, init_ch, 7, 2, 3, bias=False)]
        # Identity keeps Sequential indices (and state-dict keys) identical
        # between the fused and plain variants
        layers += [stem_bn, nn.Identity()] if stem_bn else             [nn.BatchNorm2d(init_ch), nn.ReLU(inplace=True)]
        layers.append(nn.MaxPool2d(3, 2, 1))
        ch = init_ch
        for i, n in enumerate(block_cfg):
            for _ in range(n):
                layers.append(DenseLayer(ch, growth, fused_bn=fused_bn))
                ch += growth
            if i != len(block_cfg) - 1:
                layers.append(Transition(ch, ch // 2, fused_bn=fused_bn))
                ch //= 2
        tail_bn = _bn_relu(ch, fused_bn)
        layers += [tail_bn, nn.Identity()] if tail_bn else             [nn.BatchNorm2d(ch), nn.ReLU(inplace=True)]
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Linear(ch, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight)
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def forward(self, x):
        x = F.adaptive_avg_pool2d(self.features(x), 1).flatten(1)
        return self.classifier(x)


def densenet121(num_classes=1000, fused_bn=False):
    return DenseNet([6, 12, 24, 16], num_classes=num_classes,
                    fused_bn=fused_bn)


def densenet169(num_classes=1000, fused_bn=False):
    return DenseNet([6, 12, 32, 32], num_classes=num_classes,
                    fused_bn=fused_bn)


def densenet201(num_classes=1000, fused_bn=False):
    return DenseNet([6, 12, 48, 32], num_classes=num_classes,
                    fused_bn=fused_bn)
