# Copyright (c) Flashy-AMD authors.
"""ResNet-18/50 on the native NHWC gfx950 kernel path.

Same architecture as :mod:`flashy_amd.models.resnet` but built from
``flashy_amd.nn.Conv2d`` / ``BatchNorm2d``: bf16 NHWC activations, MFMA
implicit-GEMM convs, fused BN(+residual)+ReLU, weight grads accumulated
straight into the flat fp32 optimizer buffers.  GPU-only.

``from_torch`` copies weights from the torch-module twin (for numerics
tests): conv [K,C,R,S] -> [K,R,S,C] permute.
"""
from __future__ import annotations

import typing as tp

import torch
from torch import nn

from .. import nn as fnn


class NativeBasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 downsample: bool = False):
        super().__init__()
        self.conv1 = fnn.Conv2d(cin, planes, 3, stride, 1, feeds_bn=True)
        self.bn1 = fnn.BatchNorm2d(planes)
        self.conv2 = fnn.Conv2d(planes, planes, 3, 1, 1, feeds_bn=True)
        self.bn2 = fnn.BatchNorm2d(planes)
        if downsample:
            self.dconv = fnn.Conv2d(cin, planes, 1, stride, 0, feeds_bn=True)
            self.dbn = fnn.BatchNorm2d(planes)
        else:
            self.dconv = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.dconv is not None:
            identity = self.dbn(self.dconv(x))
        else:
            identity = x
        out = self.bn1(self.conv1(x), relu=True)
        out = self.conv2(out)
        return self.bn2(out, res=identity, relu=True)


class NativeBottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, planes: int, stride: int = 1,
                 downsample: bool = False):
        super().__init__()
        self.conv1 = fnn.Conv2d(cin, planes, 1, feeds_bn=True)
        self.bn1 = fnn.BatchNorm2d(planes)
        self.conv2 = fnn.Conv2d(planes, planes, 3, stride, 1, feeds_bn=True)
        self.bn2 = fnn.BatchNorm2d(planes)
        self.conv3 = fnn.Conv2d(planes, planes * 4, 1, feeds_bn=True)
        self.bn3 = fnn.BatchNorm2d(planes * 4)
        if downsample:
            self.dconv = fnn.Conv2d(cin, planes * 4, 1, stride, 0, feeds_bn=True)
            self.dbn = fnn.BatchNorm2d(planes * 4)
        else:
            self.dconv = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.dconv is not None:
            identity = self.dbn(self.dconv(x))
        else:
            identity = x
        out = self.bn1(self.conv1(x), relu=True)
        out = self.bn2(self.conv2(out), relu=True)
        out = self.conv3(out)
        return self.bn3(out, res=identity, relu=True)


class NativeResNet(nn.Module):
    """ResNet on the native NHWC kernels; input [N, 3, H, W] (NCHW — permuted
    to NHWC once).  ``imagenet_stem`` uses the 7x7/s2 conv + 3x3/s2 maxpool
    stem (224-class inputs, BASELINE config 5); default is the CIFAR stem."""

    def __init__(self, block: type, layers: tp.Sequence[int],
                 num_classes: int = 10, imagenet_stem: bool = False):
        super().__init__()
        self.imagenet_stem = imagenet_stem
        if imagenet_stem:
            self.stem_conv = fnn.Conv2d(3, 64, 7, 2, 3, input_grad=False)
            self.stem_pool = fnn.MaxPool2d(3, 2, 1)
        else:
            self.stem_conv = fnn.Conv2d(3, 64, 3, 1, 1, input_grad=False)
            self.stem_pool = None
        self.stem_bn = fnn.BatchNorm2d(64)
        self.inplanes = 64
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.fc = fnn.Linear(512 * block.expansion, num_classes)
        self._wt_cache = None

    def _make_layer(self, block: type, planes: int, n: int, stride: int = 1):
        downsample = stride != 1 or self.inplanes != planes * block.expansion
        blocks = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        blocks += [block(self.inplanes, planes) for _ in range(1, n)]
        return nn.Sequential(*blocks)

    def enable_wt_cache(self) -> "NativeResNet":
        """With a bf16-mirror flat optimizer attached: refresh every conv's
        RSCK weight copy with ONE kernel per step (instead of one transpose
        launch per conv per backward).  Call after the optimizer exists."""
        cache = fnn.WtCache(self)
        self._wt_cache = cache if cache.active else None
        return self

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._wt_cache is not None and torch.is_grad_enabled():
            self._wt_cache.refresh()
        if x.shape[1] == 3:  # NCHW input -> logical NHWC
            x = x.permute(0, 2, 3, 1).contiguous()
        if x.dtype != torch.bfloat16:
            x = x.to(torch.bfloat16)
        x = self.stem_bn(self.stem_conv(x), relu=True)
        if self.stem_pool is not None:
            x = self.stem_pool(x)
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = x.float().mean(dim=(1, 2))      # global average pool (NHWC)
        return self.fc(x)

    # -- weight import from the torch twin (numerics tests) ----------------
    @torch.no_grad()
    def from_torch(self, twin: nn.Module) -> "NativeResNet":
        tsd = twin.state_dict()
        mapping = self._torch_key_map()
        for ours, theirs in mapping.items():
            dst = dict(self.state_dict())[ours]
            src = tsd[theirs]
            if src.dim() == 4:  # conv [K,C,R,S] -> [K,R,S,C]
                src = src.permute(0, 2, 3, 1).contiguous()
            dst.copy_(src)
        return self

    def _torch_key_map(self) -> tp.Dict[str, str]:
        m = {"stem_conv.weight": "stem.0.weight",
             "stem_bn.weight": "stem.1.weight",
             "stem_bn.bias": "stem.1.bias",
             "stem_bn.running_mean": "stem.1.running_mean",
             "stem_bn.running_var": "stem.1.running_var",
             "fc.weight": "fc.weight", "fc.bias": "fc.bias"}
        for name, mod in self.named_modules():
            if not isinstance(mod, (NativeBasicBlock, NativeBottleneck)):
                continue
            n_convs = 3 if isinstance(mod, NativeBottleneck) else 2
            for i in range(1, n_convs + 1):
                m[f"{name}.conv{i}.weight"] = f"{name}.conv{i}.weight"
                for suf in ("weight", "bias", "running_mean", "running_var"):
                    m[f"{name}.bn{i}.{suf}"] = f"{name}.bn{i}.{suf}"
            if mod.dconv is not None:
                m[f"{name}.dconv.weight"] = f"{name}.downsample.0.weight"
                for suf in ("weight", "bias", "running_mean", "running_var"):
                    m[f"{name}.dbn.{suf}"] = f"{name}.downsample.1.{suf}"
        return m


def native_resnet18(num_classes: int = 10,
                    imagenet_stem: bool = False) -> NativeResNet:
    return NativeResNet(NativeBasicBlock, [2, 2, 2, 2], num_classes,
                        imagenet_stem)


def native_resnet50(num_classes: int = 10,
                    imagenet_stem: bool = False) -> NativeResNet:
    return NativeResNet(NativeBottleneck, [3, 4, 6, 3], num_classes,
                        imagenet_stem)
