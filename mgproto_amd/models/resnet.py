"""Headless ResNet feature extractors.

Re-implementation of the reference's backbone contract
(``/root/reference/models/resnet_features.py``): a ResNet trunk with the
classifier head removed, the **initial 3x3 max-pool disabled** (reference
``models/resnet_features.py:199``) so the output stride is 16 instead of 32,
per-layer ``conv_info()`` for receptive-field math, and a nonstandard
ResNet-50 with a ``[3, 4, 6, 4]`` block layout (reference
``models/resnet_features.py:276``, matching the BBN-iNaturalist checkpoint it
ships with).

Differences from the reference (deliberate):
* ``conv_info()`` describes the layers the forward pass actually runs — the
  reference counts the *skipped* max-pool in its conv-info lists
  (``models/resnet_features.py:140-142`` vs ``:199``), which makes its RF math
  believe the latent grid is half its true size.
* ``pretrained=True`` loads from a local ``pretrained_models/`` directory when
  present (there is no network access in this environment); otherwise it
  falls back to random init with a warning.

Module/parameter naming matches torchvision ResNet so reference checkpoints
load directly.
"""

import os
import warnings

import torch
import torch.nn as nn

from .fused_bn import bn_act
from .conv1x1 import GemmConv2d

PRETRAINED_DIR = os.environ.get('MGPROTO_PRETRAINED_DIR', './pretrained_models')


def conv3x3(in_planes, out_planes, stride=1):
    return nn.Conv2d(in_planes, out_planes, kernel_size=3, stride=stride,
                     padding=1, bias=False)


def conv1x1(in_planes, out_planes, stride=1):
    # stride-1 instances take the hipBLASLt GEMM path (models/conv1x1.py)
    return GemmConv2d(in_planes, out_planes, kernel_size=1, stride=stride,
                      bias=False)


class BasicBlock(nn.Module):
    expansion = 1
    num_layers = 2

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = nn.BatchNorm2d(planes)
        self.relu = nn.ReLU(inplace=True)
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = nn.BatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        # bn+relu and bn+add+relu run as single fused NHWC kernels on GPU
        # (models/fused_bn.py); the fallback path is numerically identical
        out = bn_act(self.conv1(x), self.bn1, relu=True)
        out = self.conv2(out)
        if self.downsample is not None:
            identity = bn_act(self.downsample[0](x), self.downsample[1])
        else:
            identity = x
        return bn_act(out, self.bn2, relu=True, residual=identity)

    def block_conv_info(self):
        return [3, 3], [self.stride, 1], [1, 1]


class Bottleneck(nn.Module):
    expansion = 4
    num_layers = 3

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv1x1(inplanes, planes)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = conv1x1(planes, planes * self.expansion)
        self.bn3 = nn.BatchNorm2d(planes * self.expansion)
        self.relu = nn.ReLU(inplace=True)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        out = bn_act(self.conv1(x), self.bn1, relu=True)
        out = bn_act(self.conv2(out), self.bn2, relu=True)
        out = self.conv3(out)
        if self.downsample is not None:
            identity = bn_act(self.downsample[0](x), self.downsample[1])
        else:
            identity = x
        return bn_act(out, self.bn3, relu=True, residual=identity)

    def block_conv_info(self):
        return [1, 3, 1], [1, self.stride, 1], [0, 1, 0]


class ResNetFeatures(nn.Module):
    """ResNet trunk without avg-pool/fc; initial max-pool disabled (stride 16)."""

    def __init__(self, block, layers, zero_init_residual=False):
        super().__init__()
        self.inplanes = 64

        self.conv1 = nn.Conv2d(3, 64, kernel_size=7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        # conv-info for the stem: just the conv (the max-pool is not run)
        self.kernel_sizes = [7]
        self.strides = [2]
        self.paddings = [3]

        self.block = block
        self.layers = layers
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)

        self.out_channels = 512 * block.expansion

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out', nonlinearity='relu')
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

        if zero_init_residual:
            for m in self.modules():
                if isinstance(m, Bottleneck):
                    nn.init.constant_(m.bn3.weight, 0)
                elif isinstance(m, BasicBlock):
                    nn.init.constant_(m.bn2.weight, 0)

    def _make_layer(self, block, planes, num_blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                nn.BatchNorm2d(planes * block.expansion),
            )
        blocks = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        for _ in range(1, num_blocks):
            blocks.append(block(self.inplanes, planes))

        for b in blocks:
            ks, ss, ps = b.block_conv_info()
            self.kernel_sizes.extend(ks)
            self.strides.extend(ss)
            self.paddings.extend(ps)

        return nn.Sequential(*blocks)

    def forward(self, x):
        x = bn_act(self.conv1(x), self.bn1, relu=True)
        # NOTE: no max-pool here — output stride is 16 (reference behaviour)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        return x

    def conv_info(self):
        return self.kernel_sizes, self.strides, self.paddings

    def num_layers(self):
        return self.block.num_layers * sum(self.layers) + 1

    def __repr__(self):
        return 'resnet{}_features'.format(self.num_layers() + 1)


def _try_load_local(model, filename, drop_keys=('fc.weight', 'fc.bias'),
                    key_map=None, strict=False):
    path = os.path.join(PRETRAINED_DIR, filename)
    if not os.path.isfile(path):
        warnings.warn(
            f'pretrained weights not found at {path}; using random init '
            '(no network access in this environment)')
        return model
    sd = torch.load(path, map_location='cpu', weights_only=False)
    if isinstance(sd, dict) and 'state_dict' in sd:
        sd = sd['state_dict']
    for k in drop_keys:
        sd.pop(k, None)
    if key_map is not None:
        sd = key_map(sd)
    model.load_state_dict(sd, strict=strict)
    return model


def resnet18_features(pretrained=False, **kwargs):
    model = ResNetFeatures(BasicBlock, [2, 2, 2, 2], **kwargs)
    if pretrained:
        _try_load_local(model, 'resnet18.pth')
    return model


def resnet34_features(pretrained=False, **kwargs):
    model = ResNetFeatures(BasicBlock, [3, 4, 6, 3], **kwargs)
    if pretrained:
        _try_load_local(model, 'resnet34.pth')
    return model


def _inat_key_map(sd):
    """Remap BBN-iNaturalist checkpoint keys (reference resnet_features.py:283-287)."""
    out = {}
    for k, v in sd.items():
        nk = (k.replace('module.backbone.', '')
               .replace('cb_block', 'layer4.2')
               .replace('rb_block', 'layer4.3'))
        out[nk] = v
    out.pop('module.classifier.weight', None)
    out.pop('module.classifier.bias', None)
    return out


def resnet50_features(pretrained=False, inat=True, **kwargs):
    """Nonstandard R50 ([3,4,6,4]) matching the reference's iNat checkpoint."""
    model = ResNetFeatures(Bottleneck, [3, 4, 6, 4], **kwargs)
    if pretrained:
        name = 'BBN.iNaturalist2017.res50.90epoch.best_model.pth' if inat else 'resnet50.pth'
        _try_load_local(model, name,
                        drop_keys=('module.classifier.weight', 'module.classifier.bias'),
                        key_map=_inat_key_map if inat else None)
    return model


def resnet101_features(pretrained=False, **kwargs):
    model = ResNetFeatures(Bottleneck, [3, 4, 23, 3], **kwargs)
    if pretrained:
        _try_load_local(model, 'resnet101.pth')
    return model


def resnet152_features(pretrained=False, **kwargs):
    model = ResNetFeatures(Bottleneck, [3, 8, 36, 3], **kwargs)
    if pretrained:
        _try_load_local(model, 'resnet152.pth')
    return model
