"""ResNet family, written from scratch (no torchvision in this image).

BASELINE config 4: ResNet-50 backbone under epl.replicate(world) + a
100k-class classifier under epl.split(world) — the classifier becomes a
ColumnParallelLinear and the loss a sharded-vocab softmax CE, with the
batch gathered by the Replica2Split bridge (reference example:
docs/en/api/api_examples.md:40-53)."""

import torch
import torch.nn as nn

import easyparallellibrary_amd as epl


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)
        self.down = None
        if stride != 1 or in_ch != out_ch:
            self.down = nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_ch))

    def forward(self, x):
        idn = x if self.down is None else self.down(x)
        x = self.relu(self.bn1(self.conv1(x)))
        x = self.relu(self.bn2(self.conv2(x)))
        x = self.bn3(self.conv3(x))
        return self.relu(x + idn)


class ResNetBackbone(nn.Module):
    def __init__(self, layers=(3, 4, 6, 3), width=64):
        super().__init__()
        self.stem = nn.Sequential(
            nn.Conv2d(3, width, 7, stride=2, padding=3, bias=False),
            nn.BatchNorm2d(width), nn.ReLU(inplace=True),
            nn.MaxPool2d(3, stride=2, padding=1))
        chans = [width, width * 2, width * 4, width * 8]
        blocks = []
        in_ch = width
        for i, (n, w) in enumerate(zip(layers, chans)):
            for j in range(n):
                stride = 2 if (i > 0 and j == 0) else 1
                blocks.append(Bottleneck(in_ch, w, stride))
                in_ch = w * Bottleneck.expansion
        self.blocks = nn.Sequential(*blocks)
        self.pool = nn.AdaptiveAvgPool2d(1)
        self.out_features = in_ch

    def forward(self, x):
        x = self.stem(x)
        x = self.blocks(x)
        return self.pool(x).flatten(1)


class ResNetClassifier(nn.Module):
    def __init__(self, backbone, head):
        super().__init__()
        self.backbone = backbone
        self.head = head

    def forward(self, x):
        return self.head(self.backbone(x))


def build_resnet50_split_classifier(world, num_classes=100000):
    """The BASELINE config-4 model: DP backbone + TP classifier."""
    with epl.replicate(world, name="backbone"):
        backbone = ResNetBackbone()
    with epl.split(world, name="classifier"):
        head = nn.Linear(backbone.out_features, num_classes)
    return ResNetClassifier(backbone, head)


def build_resnet50(num_classes=1000):
    with epl.replicate(device_count=1):
        backbone = ResNetBackbone()
        head = nn.Linear(backbone.out_features, num_classes)
    return ResNetClassifier(backbone, head)


def synthetic_image_batch(batch, num_classes=1000, size=224, device="cpu",
                          seed=None):
    g = torch.Generator(device="cpu")
    if seed is not None:
        g.manual_seed(seed)
    x = torch.randn(batch, 3, size, size, generator=g)
    y = torch.randint(0, num_classes, (batch,), generator=g)
    return x.to(device), y.to(device)
