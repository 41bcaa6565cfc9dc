"""DenseNet-121/169/201 (reference matrix row: densenet201 bs32)."""
import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["densenet121", "densenet169", "densenet201"]


class DenseLayer(nn.Module):
    def __init__(self, cin, growth, bn_size=4):
        super().__init__()
        self.norm1 = nn.BatchNorm2d(cin)
        self.conv1 = nn.Conv2d(cin, bn_size * growth, 1, bias=False)
        self.norm2 = nn.BatchNorm2d(bn_size * growth)
        self.conv2 = nn.Conv2d(bn_size * growth, growth, 3, padding=1,
                               bias=False)

    def forward(self, x):
        out = self.conv1(F.relu(self.norm1(x)))
        out = self.conv2(F.relu(self.norm2(out)))
        return torch.cat([x, out], 1)


class Transition(nn.Module):
    def __init__(self, cin, cout):
        super().__init__()
        self.norm = nn.BatchNorm2d(cin)
        self.conv = nn.Conv2d(cin, cout, 1, bias=False)

    def forward(self, x):
        return F.avg_pool2d(self.conv(F.relu(self.norm(x))), 2)


class DenseNet(nn.Module):
    def __init__(self, block_cfg, growth=32, init_ch=64, num_classes=1000):
        super().__init__()
        layers = [nn.Conv2d(3, init_ch, 7, 2, 3, bias=False),
                  nn.BatchNorm2d(init_ch), nn.ReLU(inplace=True),
                  nn.MaxPool2d(3, 2, 1)]
        ch = init_ch
        for i, n in enumerate(block_cfg):
            for _ in range(n):
                layers.append(DenseLayer(ch, growth))
                ch += growth
            if i != len(block_cfg) - 1:
                layers.append(Transition(ch, ch // 2))
                ch //= 2
        layers += [nn.BatchNorm2d(ch), nn.ReLU(inplace=True)]
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


def densenet121(num_classes=1000):
    return DenseNet([6, 12, 24, 16], num_classes=num_classes)


def densenet169(num_classes=1000):
    return DenseNet([6, 12, 32, 32], num_classes=num_classes)


def densenet201(num_classes=1000):
    return DenseNet([6, 12, 48, 32], num_classes=num_classes)
