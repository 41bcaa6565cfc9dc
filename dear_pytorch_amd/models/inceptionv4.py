"""InceptionV4 (the reference carries its own copy since torchvision lacked it:
*/inceptionv4.py per SURVEY.md C18).  Standard Szegedy et al. 2016 topology,
implemented from the paper's block structure."""
import torch
import torch.nn as nn

__all__ = ["inceptionv4"]


_FUSED_BN = False  # set by inceptionv4(fused_bn=...) during construction


class ConvBN(nn.Module):
    def __init__(self, cin, cout, k, stride=1, padding=0):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride, padding, bias=False)
        if _FUSED_BN and cout % 4 == 0:
            from ..ops.fused_bn import FusedBNAct2d
            self.bn = FusedBNAct2d(cout, relu=True, eps=1e-3)
            self.act = nn.Identity()
        else:
            self.bn = nn.BatchNorm2d(cout, eps=1e-3)
            self.act = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.act(self.bn(self.conv(x)))


class Stem(nn.Module):
    def __init__(self):
        super().__init__()
        self.a = nn.Sequential(ConvBN(3, 32, 3, 2), ConvBN(32, 32, 3),
                               ConvBN(32, 64, 3, padding=1))
        self.b_pool = nn.MaxPool2d(3, 2)
        self.b_conv = ConvBN(64, 96, 3, 2)
        self.c1 = nn.Sequential(ConvBN(160, 64, 1), ConvBN(64, 96, 3))
        self.c2 = nn.Sequential(ConvBN(160, 64, 1),
                                ConvBN(64, 64, (7, 1), padding=(3, 0)),
                                ConvBN(64, 64, (1, 7), padding=(0, 3)),
                                ConvBN(64, 96, 3))
        self.d_conv = ConvBN(192, 192, 3, 2)
        self.d_pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        x = self.a(x)
        x = torch.cat([self.b_pool(x), self.b_conv(x)], 1)
        x = torch.cat([self.c1(x), self.c2(x)], 1)
        return torch.cat([self.d_conv(x), self.d_pool(x)], 1)


class InceptionA(nn.Module):
    def __init__(self):
        super().__init__()
        self.b0 = ConvBN(384, 96, 1)
        self.b1 = nn.Sequential(ConvBN(384, 64, 1),
                                ConvBN(64, 96, 3, padding=1))
        self.b2 = nn.Sequential(ConvBN(384, 64, 1),
                                ConvBN(64, 96, 3, padding=1),
                                ConvBN(96, 96, 3, padding=1))
        self.b3 = nn.Sequential(nn.AvgPool2d(3, 1, 1, count_include_pad=False),
                                ConvBN(384, 96, 1))

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x), self.b3(x)], 1)


class ReductionA(nn.Module):
    def __init__(self):
        super().__init__()
        self.b0 = ConvBN(384, 384, 3, 2)
        self.b1 = nn.Sequential(ConvBN(384, 192, 1),
                                ConvBN(192, 224, 3, padding=1),
                                ConvBN(224, 256, 3, 2))
        self.pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.pool(x)], 1)


class InceptionB(nn.Module):
    def __init__(self):
        super().__init__()
        self.b0 = ConvBN(1024, 384, 1)
        self.b1 = nn.Sequential(ConvBN(1024, 192, 1),
                                ConvBN(192, 224, (1, 7), padding=(0, 3)),
                                ConvBN(224, 256, (7, 1), padding=(3, 0)))
        self.b2 = nn.Sequential(ConvBN(1024, 192, 1),
                                ConvBN(192, 192, (7, 1), padding=(3, 0)),
                                ConvBN(192, 224, (1, 7), padding=(0, 3)),
                                ConvBN(224, 224, (7, 1), padding=(3, 0)),
                                ConvBN(224, 256, (1, 7), padding=(0, 3)))
        self.b3 = nn.Sequential(nn.AvgPool2d(3, 1, 1, count_include_pad=False),
                                ConvBN(1024, 128, 1))

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.b2(x), self.b3(x)], 1)


class ReductionB(nn.Module):
    def __init__(self):
        super().__init__()
        self.b0 = nn.Sequential(ConvBN(1024, 192, 1), ConvBN(192, 192, 3, 2))
        self.b1 = nn.Sequential(ConvBN(1024, 256, 1),
                                ConvBN(256, 256, (1, 7), padding=(0, 3)),
                                ConvBN(256, 320, (7, 1), padding=(3, 0)),
                                ConvBN(320, 320, 3, 2))
        self.pool = nn.MaxPool2d(3, 2)

    def forward(self, x):
        return torch.cat([self.b0(x), self.b1(x), self.pool(x)], 1)


class InceptionC(nn.Module):
    def __init__(self):
        super().__init__()
        self.b0 = ConvBN(1536, 256, 1)
        self.b1_stem = ConvBN(1536, 384, 1)
        self.b1a = ConvBN(384, 256, (1, 3), padding=(0, 1))
        self.b1b = ConvBN(384, 256, (3, 1), padding=(1, 0))
        self.b2_stem = nn.Sequential(ConvBN(1536, 384, 1),
                                     ConvBN(384, 448, (3, 1), padding=(1, 0)),
                                     ConvBN(448, 512, (1, 3), padding=(0, 1)))
        self.b2a = ConvBN(512, 256, (1, 3), padding=(0, 1))
        self.b2b = ConvBN(512, 256, (3, 1), padding=(1, 0))
        self.b3 = nn.Sequential(nn.AvgPool2d(3, 1, 1, count_include_pad=False),
                                ConvBN(1536, 256, 1))

    def forward(self, x):
        b1 = self.b1_stem(x)
        b2 = self.b2_stem(x)
        return torch.cat([self.b0(x), self.b1a(b1), self.b1b(b1),
                          self.b2a(b2), self.b2b(b2), self.b3(x)], 1)


class InceptionV4(nn.Module):
    def __init__(self, num_classes=1000):
        super().__init__()
        blocks = [Stem()]
        blocks += [InceptionA() for _ in range(4)]
        blocks.append(ReductionA())
        blocks += [InceptionB() for _ in range(7)]
        blocks.append(ReductionB())
        blocks += [InceptionC() for _ in range(3)]
        self.features = nn.Sequential(*blocks)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.dropout = nn.Dropout(0.2)
        self.classifier = nn.Linear(1536, num_classes)

    def forward(self, x):
        x = self.avgpool(self.features(x)).flatten(1)
        return self.classifier(self.dropout(x))


def inceptionv4(num_classes=1000, fused_bn=False):
    global _FUSED_BN
    _FUSED_BN = fused_bn
    try:
        return InceptionV4(num_classes)
    finally:
        _FUSED_BN = False
