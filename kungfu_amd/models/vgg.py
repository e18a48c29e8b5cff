"""VGG16 (ImageNet-shaped) — benchmark fixture (the reference benchmarks
ResNet-50/VGG16/InceptionV3; README.md:201-211)."""
import torch.nn as nn

_CFG16 = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class VGG(nn.Module):
    def __init__(self, cfg, classes=1000):
        super().__init__()
        layers = []
        cin = 3
        for v in cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(cin, v, 3, padding=1),
                           nn.ReLU(inplace=True)]
                cin = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(inplace=True),
            nn.Dropout(),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(),
            nn.Linear(4096, classes),
        )

    def forward(self, x):
        return self.classifier(self.features(x).flatten(1))


def vgg16(classes=1000):
    return VGG(_CFG16, classes=classes)
