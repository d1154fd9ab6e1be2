"""ai-benchmark model zoo (PyTorch-ROCm, hand-written — no torchvision).

The reference's published benchmark is the ai-benchmark suite on TF 2.4.1
(/root/reference/benchmarks/ai-benchmark/Dockerfile:1-13, README.md:243-256);
these are the same 10 cases re-implemented for PyTorch-ROCm with synthetic
data and random-init weights (no network for datasets/checkpoints):

  1.1/1.2  ResNet-V2-50  inference b50 @346^2 / training b20 @346^2
  2.1/2.2  ResNet-V2-152 inference b10 @256^2 / training b10 @256^2
  3.1/3.2  VGG-16        inference b20 @224^2 / training b2  @224^2
  4.1/4.2  DeepLab       inference b2  @512^2 / training b1  @384^2
  5.1/5.2  LSTM          inference b100 (1024x300) / training b10

ResNet-V2 = pre-activation bottlenecks (BN-ReLU-conv), matching the
"resnet_v2" family the suite names.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, Dict, List, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# ResNet-V2 (pre-activation)
# ---------------------------------------------------------------------------
class PreActBottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1):
        super().__init__()
        out_ch = width * self.expansion
        self.bn1 = nn.BatchNorm2d(in_ch)
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1, bias=False)
        self.bn3 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.shortcut = None
        if stride != 1 or in_ch != out_ch:
            self.shortcut = nn.Conv2d(in_ch, out_ch, 1, stride=stride, bias=False)

    def forward(self, x):
        out = F.relu(self.bn1(x))
        sc = self.shortcut(out) if self.shortcut is not None else x
        out = self.conv1(out)
        out = self.conv2(F.relu(self.bn2(out)))
        out = self.conv3(F.relu(self.bn3(out)))
        return out + sc


class ResNetV2(nn.Module):
    def __init__(self, layers: List[int], num_classes=1000):
        super().__init__()
        self.stem = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.pool = nn.MaxPool2d(3, stride=2, padding=1)
        widths = [64, 128, 256, 512]
        blocks = []
        in_ch = 64
        for i, (n, w) in enumerate(zip(layers, widths)):
            for j in range(n):
                stride = 2 if (j == 0 and i > 0) else 1
                blocks.append(PreActBottleneck(in_ch, w, stride))
                in_ch = w * PreActBottleneck.expansion
        self.blocks = nn.Sequential(*blocks)
        self.bn_final = nn.BatchNorm2d(in_ch)
        self.fc = nn.Linear(in_ch, num_classes)

    def forward(self, x):
        x = self.pool(self.stem(x))
        x = self.blocks(x)
        x = F.relu(self.bn_final(x))
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def resnet_v2_50():
    return ResNetV2([3, 4, 6, 3])


def resnet_v2_152():
    return ResNetV2([3, 8, 36, 3])


# ---------------------------------------------------------------------------
# VGG-16
# ---------------------------------------------------------------------------
class VGG16(nn.Module):
    cfg = [64, 64, "M", 128, 128, "M", 256, 256, 256, "M",
           512, 512, 512, "M", 512, 512, 512, "M"]

    def __init__(self, num_classes=1000):
        super().__init__()
        layers = []
        in_ch = 3
        for v in self.cfg:
            if v == "M":
                layers.append(nn.MaxPool2d(2, 2))
            else:
                layers += [nn.Conv2d(in_ch, v, 3, padding=1), nn.ReLU(inplace=True)]
                in_ch = v
        self.features = nn.Sequential(*layers)
        self.classifier = nn.Sequential(
            nn.Linear(512 * 7 * 7, 4096), nn.ReLU(inplace=True), nn.Dropout(0.5),
            nn.Linear(4096, 4096), nn.ReLU(inplace=True), nn.Dropout(0.5),
            nn.Linear(4096, num_classes),
        )

    def forward(self, x):
        x = self.features(x)
        x = F.adaptive_avg_pool2d(x, 7).flatten(1)
        return self.classifier(x)


# ---------------------------------------------------------------------------
# DeepLabV3 (ResNet-50-V2 backbone, output stride 16, ASPP head)
# ---------------------------------------------------------------------------
class ASPP(nn.Module):
    def __init__(self, in_ch, out_ch=256):
        super().__init__()
        rates = [6, 12, 18]
        self.branches = nn.ModuleList(
            [nn.Sequential(nn.Conv2d(in_ch, out_ch, 1, bias=False),
                           nn.BatchNorm2d(out_ch), nn.ReLU(inplace=True))]
            + [nn.Sequential(
                nn.Conv2d(in_ch, out_ch, 3, padding=r, dilation=r, bias=False),
                nn.BatchNorm2d(out_ch), nn.ReLU(inplace=True))
               for r in rates])
        # image-pooling branch: GroupNorm, not BatchNorm — after global
        # pooling the spatial extent is 1x1, and the published training
        # config is batch=1 (README.md:253-254), where BatchNorm cannot
        # compute statistics (1 value/channel) and raises in train mode
        self.gp = nn.Sequential(
            nn.AdaptiveAvgPool2d(1), nn.Conv2d(in_ch, out_ch, 1, bias=False),
            nn.GroupNorm(32, out_ch), nn.ReLU(inplace=True))
        self.project = nn.Sequential(
            nn.Conv2d(out_ch * 5, out_ch, 1, bias=False),
            nn.BatchNorm2d(out_ch), nn.ReLU(inplace=True))

    def forward(self, x):
        size = x.shape[-2:]
        feats = [b(x) for b in self.branches]
        gp = F.interpolate(self.gp(x), size=size, mode="bilinear",
                           align_corners=False)
        return self.project(torch.cat(feats + [gp], dim=1))


class DeepLabV3(nn.Module):
    def __init__(self, num_classes=21):
        super().__init__()
        backbone = ResNetV2([3, 4, 6, 3])
        self.stem = backbone.stem
        self.pool = backbone.pool
        # output stride 16: keep stage strides 1,2,2 and dilate the last stage
        self.blocks = backbone.blocks
        for m in list(self.blocks)[-3:]:
            if isinstance(m, PreActBottleneck):
                m.conv2.stride = (1, 1)
                m.conv2.dilation = (2, 2)
                m.conv2.padding = (2, 2)
                if m.shortcut is not None:
                    m.shortcut.stride = (1, 1)
        self.aspp = ASPP(2048)
        self.head = nn.Conv2d(256, num_classes, 1)

    def forward(self, x):
        size = x.shape[-2:]
        h = self.pool(self.stem(x))
        h = self.blocks(h)
        h = self.aspp(h)
        h = self.head(h)
        return F.interpolate(h, size=size, mode="bilinear", align_corners=False)


# ---------------------------------------------------------------------------
# LSTM (sentiment-style: embedded sequence -> logit)
# ---------------------------------------------------------------------------
class LSTMNet(nn.Module):
    def __init__(self, input_size=300, hidden=512, layers=2, num_classes=2):
        super().__init__()
        self.lstm = nn.LSTM(input_size, hidden, num_layers=layers,
                            batch_first=True)
        self.fc = nn.Linear(hidden, num_classes)

    def forward(self, x):
        out, _ = self.lstm(x)
        return self.fc(out[:, -1])


# ---------------------------------------------------------------------------
# Case table
# ---------------------------------------------------------------------------
@dataclass
class BenchCase:
    name: str
    model_fn: Callable[[], nn.Module]
    phase: str          # "inference" | "training"
    batch: int
    input_shape: Tuple[int, ...]   # without batch dim
    num_classes: int = 1000
    seg: bool = False   # segmentation loss shape


CASES: Dict[str, BenchCase] = {
    "resnet50_inf": BenchCase("resnet50_inf", resnet_v2_50, "inference", 50, (3, 346, 346)),
    "resnet50_train": BenchCase("resnet50_train", resnet_v2_50, "training", 20, (3, 346, 346)),
    "resnet152_inf": BenchCase("resnet152_inf", resnet_v2_152, "inference", 10, (3, 256, 256)),
    "resnet152_train": BenchCase("resnet152_train", resnet_v2_152, "training", 10, (3, 256, 256)),
    "vgg16_inf": BenchCase("vgg16_inf", VGG16, "inference", 20, (3, 224, 224)),
    "vgg16_train": BenchCase("vgg16_train", VGG16, "training", 2, (3, 224, 224)),
    "deeplab_inf": BenchCase("deeplab_inf", DeepLabV3, "inference", 2, (3, 512, 512),
                             num_classes=21, seg=True),
    "deeplab_train": BenchCase("deeplab_train", DeepLabV3, "training", 1, (3, 384, 384),
                               num_classes=21, seg=True),
    "lstm_inf": BenchCase("lstm_inf", LSTMNet, "inference", 100, (1024, 300),
                          num_classes=2),
    "lstm_train": BenchCase("lstm_train", LSTMNet, "training", 10, (1024, 300),
                            num_classes=2),
}

DEFAULT_CASES = ["resnet50_inf", "resnet50_train"]


def synthetic_batch(case: BenchCase, device, dtype=torch.float32):
    x = torch.randn(case.batch, *case.input_shape, device=device, dtype=dtype)
    if case.seg:
        y = torch.randint(0, case.num_classes,
                          (case.batch, *case.input_shape[1:]), device=device)
    else:
        y = torch.randint(0, case.num_classes, (case.batch,), device=device)
    return x, y


def build(case: BenchCase, device, dtype=torch.float32) -> nn.Module:
    model = case.model_fn().to(device=device, dtype=dtype)
    if case.phase == "inference":
        model.eval()
    return model


def step(case: BenchCase, model: nn.Module, batch, optimizer=None) -> None:
    """One benchmark step: forward (inference) or fwd+loss+bwd+opt (training)."""
    x, y = batch
    if case.phase == "inference":
        with torch.no_grad():
            model(x)
    else:
        out = model(x)
        loss = F.cross_entropy(out.float(), y)
        loss.backward()
        if optimizer is not None:
            optimizer.step()
            optimizer.zero_grad(set_to_none=True)


# ---------------------------------------------------------------------------
# CLI: the in-cluster benchmark entry point used by benchmarks/ai-benchmark/*
# (the reference runs the ai-benchmark binary inside the Job; here the zoo is
# the suite).  --steps 0 loops forever (density/serving pods).
# ---------------------------------------------------------------------------
def main(argv=None):
    import argparse
    import json
    import time

    p = argparse.ArgumentParser("ai-benchmark-zoo")
    p.add_argument("--cases", default="all", help="comma list or 'all'")
    p.add_argument("--steps", type=int, default=50,
                   help="timed steps per case; 0 = run forever")
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--device", default="cuda:0")
    args = p.parse_args(argv)

    names = list(CASES) if args.cases == "all" else [
        c.strip() for c in args.cases.split(",") if c.strip()]
    dev = torch.device(args.device)
    results = {}
    for name in names:
        case = CASES[name]
        model = build(case, dev)
        batch = synthetic_batch(case, dev)
        opt = (torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
               if case.phase == "training" else None)
        for _ in range(args.warmup):
            step(case, model, batch, opt)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        if args.steps == 0:
            print(f"{name}: serving forever", flush=True)
            while True:
                step(case, model, batch, opt)
        t0 = time.perf_counter()
        for _ in range(args.steps):
            step(case, model, batch, opt)
        if dev.type == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        results[name] = case.batch * args.steps / dt
        print(f"{name}: {results[name]:.2f} samples/s", flush=True)
    print(json.dumps({"samples_per_s": results}), flush=True)


if __name__ == "__main__":
    main()
