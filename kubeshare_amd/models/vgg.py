"""VGG16 — the third workload family of the reference's distributed e2e
suite (test/distribute/**/vgg16*). Conv->BN->ReLU triples are structured
as blocks so kubeshare_amd.ops.fuse_model can route them through the
fused NHWC bf16 BN+ReLU gfx950 kernel (same op the ResNet blocks use)."""
import torch.nn as nn

_CFG16 = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
          512, 512, 512, "M", 512, 512, 512, "M"]


class ConvBNReLU(nn.Module):
    fused_ops = False  # set by kubeshare_amd.ops.fuse_model

    def __init__(self, in_ch: int, out_ch: int):
        super().__init__()
        self.conv = nn.Conv2d(in_ch, out_ch, 3, padding=1)
        self.bn = nn.BatchNorm2d(out_ch)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        if self.fused_ops:
            from .. import ops
            return ops.bn_relu(self.conv(x), self.bn)
        return self.relu(self.bn(self.conv(x)))


class VGG(nn.Module):
    def __init__(self, cfg, num_classes: int = 1000):
        super().__init__()
        layers, in_ch = [], 3
        for v in cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers.append(ConvBNReLU(in_ch, v))
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(True), nn.Dropout(),
            nn.Linear(4096, 4096), nn.ReLU(True), nn.Dropout(),
            nn.Linear(4096, num_classes))

    def forward(self, x):
        x = self.avgpool(self.features(x)).flatten(1)
        return self.classifier(x)


def vgg16(num_classes: int = 1000) -> VGG:
    return VGG(_CFG16, num_classes)
