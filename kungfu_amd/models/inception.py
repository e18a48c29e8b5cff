"""Inception-V3 (ImageNet-shaped) — third model of the reference's
benchmark suite (README.md:201-211: ResNet-50 / VGG16 / InceptionV3).

Standard Szegedy et al. 2015 architecture (no aux head; 299x299 input),
written directly — no torchvision dependency.
"""
import torch
import torch.nn as nn


class ConvBN(nn.Module):
    def __init__(self, cin, cout, **kw):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, bias=False, **kw)
        self.bn = nn.BatchNorm2d(cout, eps=0.001)

    def forward(self, x):
        return torch.relu(self.bn(self.conv(x)))


class InceptionA(nn.Module):
    def __init__(self, cin, pool):
        super().__init__()
        self.b1 = ConvBN(cin, 64, kernel_size=1)
        self.b5 = nn.Sequential(ConvBN(cin, 48, kernel_size=1),
                                ConvBN(48, 64, kernel_size=5, padding=2))
        self.b3 = nn.Sequential(ConvBN(cin, 64, kernel_size=1),
                                ConvBN(64, 96, kernel_size=3, padding=1),
                                ConvBN(96, 96, kernel_size=3, padding=1))
        self.bp = ConvBN(cin, pool, kernel_size=1)

    def forward(self, x):
        p = nn.functional.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat(
            [self.b1(x), self.b5(x), self.b3(x), self.bp(p)], 1)


class ReductionA(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b3 = ConvBN(cin, 384, kernel_size=3, stride=2)
        self.b33 = nn.Sequential(ConvBN(cin, 64, kernel_size=1),
                                 ConvBN(64, 96, kernel_size=3, padding=1),
                                 ConvBN(96, 96, kernel_size=3, stride=2))

    def forward(self, x):
        p = nn.functional.max_pool2d(x, 3, stride=2)
        return torch.cat([self.b3(x), self.b33(x), p], 1)


class InceptionB(nn.Module):
    def __init__(self, cin, c7):
        super().__init__()
        self.b1 = ConvBN(cin, 192, kernel_size=1)
        self.b7 = nn.Sequential(
            ConvBN(cin, c7, kernel_size=1),
            ConvBN(c7, c7, kernel_size=(1, 7), padding=(0, 3)),
            ConvBN(c7, 192, kernel_size=(7, 1), padding=(3, 0)))
        self.b77 = nn.Sequential(
            ConvBN(cin, c7, kernel_size=1),
            ConvBN(c7, c7, kernel_size=(7, 1), padding=(3, 0)),
            ConvBN(c7, c7, kernel_size=(1, 7), padding=(0, 3)),
            ConvBN(c7, c7, kernel_size=(7, 1), padding=(3, 0)),
            ConvBN(c7, 192, kernel_size=(1, 7), padding=(0, 3)))
        self.bp = ConvBN(cin, 192, kernel_size=1)

    def forward(self, x):
        p = nn.functional.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat(
            [self.b1(x), self.b7(x), self.b77(x), self.bp(p)], 1)


class ReductionB(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b3 = nn.Sequential(ConvBN(cin, 192, kernel_size=1),
                                ConvBN(192, 320, kernel_size=3, stride=2))
        self.b7 = nn.Sequential(
            ConvBN(cin, 192, kernel_size=1),
            ConvBN(192, 192, kernel_size=(1, 7), padding=(0, 3)),
            ConvBN(192, 192, kernel_size=(7, 1), padding=(3, 0)),
            ConvBN(192, 192, kernel_size=3, stride=2))

    def forward(self, x):
        p = nn.functional.max_pool2d(x, 3, stride=2)
        return torch.cat([self.b3(x), self.b7(x), p], 1)


class InceptionC(nn.Module):
    def __init__(self, cin):
        super().__init__()
        self.b1 = ConvBN(cin, 320, kernel_size=1)
        self.b3_stem = ConvBN(cin, 384, kernel_size=1)
        self.b3_a = ConvBN(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.b3_b = ConvBN(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.b33_stem = nn.Sequential(
            ConvBN(cin, 448, kernel_size=1),
            ConvBN(448, 384, kernel_size=3, padding=1))
        self.b33_a = ConvBN(384, 384, kernel_size=(1, 3), padding=(0, 1))
        self.b33_b = ConvBN(384, 384, kernel_size=(3, 1), padding=(1, 0))
        self.bp = ConvBN(cin, 192, kernel_size=1)

    def forward(self, x):
        s3 = self.b3_stem(x)
        s33 = self.b33_stem(x)
        p = nn.functional.avg_pool2d(x, 3, stride=1, padding=1)
        return torch.cat([
            self.b1(x), self.b3_a(s3), self.b3_b(s3), self.b33_a(s33),
            self.b33_b(s33), self.bp(p)
        ], 1)


class InceptionV3(nn.Module):
    def __init__(self, classes=1000):
        super().__init__()
        self.stem = nn.Sequential(
            ConvBN(3, 32, kernel_size=3, stride=2),
            ConvBN(32, 32, kernel_size=3),
            ConvBN(32, 64, kernel_size=3, padding=1),
            nn.MaxPool2d(3, stride=2),
            ConvBN(64, 80, kernel_size=1),
            ConvBN(80, 192, kernel_size=3),
            nn.MaxPool2d(3, stride=2),
        )
        self.blocks = nn.Sequential(
            InceptionA(192, 32), InceptionA(256, 64), InceptionA(288, 64),
            ReductionA(288),
            InceptionB(768, 128), InceptionB(768, 160),
            InceptionB(768, 160), InceptionB(768, 192),
            ReductionB(768),
            InceptionC(1280), InceptionC(2048),
        )
        self.fc = nn.Linear(2048, classes)

    def forward(self, x):
        x = self.blocks(self.stem(x))
        x = nn.functional.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def inception_v3(classes=1000):
    return InceptionV3(classes=classes)
