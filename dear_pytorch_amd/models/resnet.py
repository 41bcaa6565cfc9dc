"""ResNet family (ResNet-18/34/50/101/152), self-contained (no torchvision in
this environment).  Matches the standard ImageNet architecture the reference
benchmarks pull from torchvision (dear/imagenet_benchmark.py:69-71).

``fused_bn=True`` swaps nn.BatchNorm2d + ReLU + residual-add for the
CDNA4 fused NHWC kernel (ops/fused_bn.py) — identical numerics, same
state-dict keys, ~fewer HBM passes + kernel launches per block."""
import torch.nn as nn

__all__ = ["resnet18", "resnet34", "resnet50", "resnet101", "resnet152"]


def conv3x3(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def _bn(planes, relu, fused):
    if fused:
        from ..ops.fused_bn import FusedBNAct2d
        return FusedBNAct2d(planes, relu=relu)
    return nn.BatchNorm2d(planes)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1, down=None, fused_bn=False):
        super().__init__()
        self.fused = fused_bn
        self.conv1 = conv3x3(cin, planes, stride)
        self.bn1 = _bn(planes, True, fused_bn)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = _bn(planes, True, fused_bn)
        self.relu = nn.ReLU(inplace=True)
        self.down = down

    def forward(self, x):
        idn = self.down(x) if self.down is not None else x
        if self.fused:
            out = self.bn1(self.conv1(x))
            return self.bn2(self.conv2(out), residual=idn)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + idn)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, down=None, fused_bn=False):
        super().__init__()
        self.fused = fused_bn
        self.conv1 = nn.Conv2d(cin, planes, 1, bias=False)
        self.bn1 = _bn(planes, True, fused_bn)
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = _bn(planes, True, fused_bn)
        self.conv3 = nn.Conv2d(planes, planes * 4, 1, bias=False)
        self.bn3 = _bn(planes * 4, True, fused_bn)
        self.relu = nn.ReLU(inplace=True)
        self.down = down

    def forward(self, x):
        idn = self.down(x) if self.down is not None else x
        if self.fused:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            return self.bn3(self.conv3(out), residual=idn)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + idn)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, fused_bn=False):
        super().__init__()
        self.inplanes = 64
        self.fused_bn = fused_bn
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = _bn(64, True, fused_bn)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], 2)
        self.layer3 = self._make_layer(block, 256, layers[2], 2)
        self.layer4 = self._make_layer(block, 512, layers[3], 2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _make_layer(self, block, planes, n, stride=1):
        down = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            down = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * block.expansion, 1,
                          stride=stride, bias=False),
                _bn(planes * block.expansion, False, self.fused_bn))
        layers = [block(self.inplanes, planes, stride, down,
                        fused_bn=self.fused_bn)]
        self.inplanes = planes * block.expansion
        for _ in range(1, n):
            layers.append(block(self.inplanes, planes,
                                fused_bn=self.fused_bn))
        return nn.Sequential(*layers)

    def forward(self, x):
        if self.fused_bn:
            x = self.maxpool(self.bn1(self.conv1(x)))
        else:
            x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet18(num_classes=1000, fused_bn=False):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, fused_bn)


def resnet34(num_classes=1000, fused_bn=False):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, fused_bn)


def resnet50(num_classes=1000, fused_bn=False):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, fused_bn)


def resnet101(num_classes=1000, fused_bn=False):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, fused_bn)


def resnet152(num_classes=1000, fused_bn=False):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, fused_bn)
