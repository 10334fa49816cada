"""Retrieval backbones: SSCD-style ResNet50 (+GeM head) and VGG16.

Capability parity (torchvision is not installed here — these are
from-scratch implementations):
* SSCD: torchscript ResNet50 with GeM pooling -> 512-d L2-normed
  descriptor (/root/reference/diff_retrieval.py:277-285,
  embedding_search/utils.py:15-33). If a local torchscript file exists
  (./pretrainedmodels/sscd_*.torchscript.pt) it is loaded; otherwise a
  randomly-initialized network of the same architecture is used (no
  network in this environment; BASELINE configs run on synthetic data).
* VGG16 (features + fc2 4096-d) for Improved Precision & Recall
  (/root/reference/metrics/ipr.py:41,146-148).


MI355X note (SURVEY §2.4.F / §7 stage 5): bottleneck/downsample convs are
DcrConv2d — they dispatch to the in-tree implicit-GEMM kernel when run
bf16 + channels_last (C%32, K%64, 1x1/3x3, stride 1|2); the 7x7 C=3 stem
and fp32 eval runs stay on MIOpen, which measured 0.94-1.24x parity with
our kernel on SD shapes — for the inference-only,
amortized metric passes that parity makes a dedicated small-C stem
kernel not worth its maintenance.
"""
from __future__ import annotations

from pathlib import Path
from typing import Optional

import torch
import torch.nn as nn

from ..ops.conv import Conv2d as DcrConv2d
import torch.nn.functional as F


# --------------------------------------------------------------- ResNet50
class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = DcrConv2d(in_ch, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = DcrConv2d(planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = DcrConv2d(planes, planes * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.downsample = downsample

    def forward(self, x):
        idt = x
        out = F.relu(self.bn1(self.conv1(x)))
        out = F.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            idt = self.downsample(x)
        return F.relu(out + idt)


class ResNet50(nn.Module):
    def __init__(self, num_classes: Optional[int] = None):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, 3)
        self.layer2 = self._make_layer(128, 4, stride=2)
        self.layer3 = self._make_layer(256, 6, stride=2)
        self.layer4 = self._make_layer(512, 3, stride=2)
        self.fc = nn.Linear(2048, num_classes) if num_classes else None

    def _make_layer(self, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * 4:
            downsample = nn.Sequential(
                DcrConv2d(self.inplanes, planes * 4, 1, stride=stride, bias=False),
                nn.BatchNorm2d(planes * 4))
        layers = [Bottleneck(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * 4
        layers += [Bottleneck(self.inplanes, planes) for _ in range(blocks - 1)]
        return nn.Sequential(*layers)

    def forward_features(self, x):
        x = F.relu(self.bn1(self.conv1(x)))
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        return self.layer4(x)  # [B, 2048, H/32, W/32]

    def forward(self, x):
        f = self.forward_features(x)
        f = F.adaptive_avg_pool2d(f, 1).flatten(1)
        return self.fc(f) if self.fc is not None else f


class GeMPool(nn.Module):
    """Generalized-mean pooling (SSCD head)."""

    def __init__(self, p: float = 3.0, eps: float = 1e-6):
        super().__init__()
        self.p = p
        self.eps = eps

    def forward(self, x):
        return F.adaptive_avg_pool2d(x.clamp(min=self.eps).pow(self.p), 1) \
            .pow(1.0 / self.p).flatten(1)


class SSCDModel(nn.Module):
    """ResNet50 trunk + GeM + 512-d projection + L2 norm (SSCD descriptor)."""

    def __init__(self, dims: int = 512):
        super().__init__()
        self.backbone = ResNet50()
        self.pool = GeMPool(3.0)
        self.embeddings = nn.Linear(2048, dims)
        self.dims = dims

    def forward(self, x):
        f = self.backbone.forward_features(x)
        f = self.pool(f)
        f = self.embeddings(f)
        return F.normalize(f, dim=-1)


_SSCD_FILES = {
    "sscd": "sscd_disc_mixup.torchscript.pt",
    "sscd_im": "sscd_imagenet_mixup.torchscript.pt",
    "sscd_disc_large": "sscd_disc_large.torchscript.pt",
}


def load_sscd(pt_style: str = "sscd", model_dir: str = "./pretrainedmodels",
              device: str | torch.device = "cpu"):
    """torchscript weights if present (reference dir naming,
    diff_retrieval.py:279-283), else random-init SSCDModel."""
    fname = _SSCD_FILES.get(pt_style, _SSCD_FILES["sscd"])
    for d in (model_dir, "./pretrainedmodels", "./pretrained_models"):
        p = Path(d) / fname
        if p.exists():
            return torch.jit.load(str(p), map_location=device).eval()
    m = SSCDModel(512 if pt_style != "sscd_disc_large" else 1024)
    torch.manual_seed(0)  # deterministic random backbone across processes
    for mod in m.modules():
        if isinstance(mod, (nn.Conv2d, nn.Linear)):
            nn.init.kaiming_normal_(mod.weight)
    return m.to(device).eval()


# --------------------------------------------------------------- VGG16
class VGG16(nn.Module):
    """VGG16 with the torchvision layer layout (features/classifier)."""

    CFG = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
           512, 512, 512, "M", 512, 512, 512, "M"]

    def __init__(self, num_classes: int = 1000,
                 weights_path: str | None = None):
        super().__init__()
        layers, in_ch = [], 3
        for v in self.CFG:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [DcrConv2d(in_ch, v, 3, padding=1), nn.ReLU(inplace=True)]
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.avgpool = nn.AdaptiveAvgPool2d(7)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(True), nn.Dropout(),
            nn.Linear(4096, 4096), nn.ReLU(True), nn.Dropout(),
            nn.Linear(4096, num_classes))
        # torchvision-format ImageNet weights when available offline
        # (reference metrics/ipr.py:41 downloads them via torchvision)
        wp = Path(weights_path) if weights_path else \
            Path("./pretrainedmodels") / "vgg16.pth"
        if wp.exists():
            sd = torch.load(str(wp), map_location="cpu", weights_only=True)
            self.load_state_dict(sd, strict=False)

    def forward(self, x):
        x = self.features(x)
        x = self.avgpool(x).flatten(1)
        return self.classifier(x)

    def fc2_features(self, x):
        """4096-d fc2 activations (IPR feature space, metrics/ipr.py:146)."""
        x = self.features(x)
        x = self.avgpool(x).flatten(1)
        for layer in list(self.classifier)[:5]:  # up to and incl. fc2+ReLU
            x = layer(x)
        return x
