"""ResNet-50 (v1.5, ImageNet-shaped) — the flagship benchmark model.

Standard bottleneck ResNet (He et al. 2015), written here directly so the
benchmark has no torchvision dependency; the reference benchmarks build the
same architecture from tf.keras.applications
(benchmarks/system/benchmark_kungfu.py:36-48). Parameter census matches the
canonical model: 161 trainable tensors / ~25.6M params
(tests/go/fakemodel/resnet50-imagenet.go).
"""
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, stride=1, downsample=None,
                 fused_bn=False):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = nn.Conv2d(cin, width, 1, bias=False)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.conv3 = nn.Conv2d(width, cout, 1, bias=False)
        self.fused = fused_bn
        if fused_bn:
            from kungfu_amd.ops.fused_bn import FusedBNReLU2d

            # bn+relu fused; bn3 additionally fuses the residual add+relu
            self.bn1 = FusedBNReLU2d(width, relu=True)
            self.bn2 = FusedBNReLU2d(width, relu=True)
            self.bn3 = FusedBNReLU2d(cout, relu=True)
        else:
            self.bn1 = nn.BatchNorm2d(width)
            self.bn2 = nn.BatchNorm2d(width)
            self.bn3 = nn.BatchNorm2d(cout)
            self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idt = x
        if self.downsample is not None:
            idt = self.downsample(x)
        if self.fused:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            return self.bn3(self.conv3(out), residual=idt)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + idt)


class ResNet(nn.Module):
    def __init__(self, layers, classes=1000, fused_bn=False):
        super().__init__()
        self.cin = 64
        self.fused = fused_bn
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        if fused_bn:
            from kungfu_amd.ops.fused_bn import FusedBNReLU2d

            self.bn1 = FusedBNReLU2d(64, relu=True)
        else:
            self.bn1 = nn.BatchNorm2d(64)
            self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._stage(64, layers[0])
        self.layer2 = self._stage(128, layers[1], stride=2)
        self.layer3 = self._stage(256, layers[2], stride=2)
        self.layer4 = self._stage(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * Bottleneck.expansion, classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.ones_(m.weight)
                nn.init.zeros_(m.bias)

    def _stage(self, width, blocks, stride=1):
        downsample = None
        cout = width * Bottleneck.expansion
        if stride != 1 or self.cin != cout:
            if self.fused:
                from kungfu_amd.ops.fused_bn import FusedBNReLU2d

                ds_bn = FusedBNReLU2d(cout, relu=False)
            else:
                ds_bn = nn.BatchNorm2d(cout)
            downsample = nn.Sequential(
                nn.Conv2d(self.cin, cout, 1, stride=stride, bias=False),
                ds_bn,
            )
        layers = [Bottleneck(self.cin, width, stride, downsample,
                             fused_bn=self.fused)]
        self.cin = cout
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.cin, width,
                                     fused_bn=self.fused))
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.bn1(self.conv1(x)) if self.fused else
                         self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.fc(self.avgpool(x).flatten(1))


def resnet50(classes=1000, fused_bn=False):
    return ResNet([3, 4, 6, 3], classes=classes, fused_bn=fused_bn)
