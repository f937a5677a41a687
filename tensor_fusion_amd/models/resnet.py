"""ResNet-50 — the fractional-vGPU validation workload.

BASELINE config 2: local fractional vGPU on one MI355X, 25 % TFLOPS /
8 GB VRAM limit, bf16 inference. torchvision is not in the image, so
this is a from-scratch ResNet-50 (standard bottleneck v1.5); random
init, synthetic NCHW batches.

Run one benchmark child process:
    python -m tensor_fusion_amd.models.resnet --batch 64 --steps 32
prints {"img_s": ..., "ms_per_step": ...}.
"""
from __future__ import annotations

import argparse
import json
import time

import torch
import torch.nn as nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(cin, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample

    def forward(self, x):
        idn = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            idn = self.downsample(x)
        return self.relu(out + idn)


class ResNet50(nn.Module):
    LAYERS = (3, 4, 6, 3)

    def __init__(self, num_classes: int = 1000):
        super().__init__()
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, self.LAYERS[0])
        self.layer2 = self._make_layer(128, self.LAYERS[1], stride=2)
        self.layer3 = self._make_layer(256, self.LAYERS[2], stride=2)
        self.layer4 = self._make_layer(512, self.LAYERS[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * 4, num_classes)

    def _make_layer(self, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * 4:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * 4, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(planes * 4))
        layers = [Bottleneck(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * 4
        layers += [Bottleneck(self.inplanes, planes)
                   for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


@torch.no_grad()
def infer_bench(batch: int, steps: int, warmup: int, device="cuda",
                dtype=torch.bfloat16):
    model = ResNet50().to(device=device, dtype=dtype).eval()
    x = torch.randn(batch, 3, 224, 224, device=device, dtype=dtype)
    for _ in range(warmup):
        model(x)
    if device != "cpu":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        model(x)
    if device != "cpu":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return batch * steps / dt, dt / steps * 1000.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--steps", type=int, default=32)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--device", default="cuda")
    args = ap.parse_args()
    img_s, ms = infer_bench(args.batch, args.steps, args.warmup,
                            device=args.device)
    print(json.dumps({"img_s": img_s, "ms_per_step": ms,
                      "batch": args.batch}), flush=True)


if __name__ == "__main__":
    main()
