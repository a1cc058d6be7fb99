"""Headless DenseNet feature extractors.

Re-implementation of the reference backbone contract
(``/root/reference/models/densenet_features.py``): DenseNet-BC trunk without
the classifier, the **initial max-pool disabled** (reference
``models/densenet_features.py:116``) so output stride is 16, a final
``norm5`` + ReLU, and per-layer ``conv_info()``.

As with the ResNet trunk, ``conv_info()`` here describes the actual forward
(the reference counts the skipped pool in its lists). Module naming matches
torchvision DenseNet so reference checkpoints load directly.
"""

import os
import re
import warnings
from collections import OrderedDict

import torch
import torch.nn as nn
import torch.nn.functional as F

from .resnet import PRETRAINED_DIR
from .fused_bn import FusedBatchNorm2d
from .conv1x1 import GemmConv2d


class _DenseLayer(nn.Sequential):
    num_layers = 2

    def __init__(self, num_input_features, growth_rate, bn_size, drop_rate):
        super().__init__()
        self.add_module('norm1', FusedBatchNorm2d(num_input_features,
                                                  fused_relu=True))
        self.add_module('relu1', nn.Identity())
        self.add_module('conv1', GemmConv2d(num_input_features,
                                            bn_size * growth_rate,
                                            kernel_size=1, stride=1,
                                            bias=False))
        self.add_module('norm2', FusedBatchNorm2d(bn_size * growth_rate,
                                                  fused_relu=True))
        self.add_module('relu2', nn.Identity())
        self.add_module('conv2', nn.Conv2d(bn_size * growth_rate, growth_rate,
                                           kernel_size=3, stride=1, padding=1, bias=False))
        self.drop_rate = drop_rate

    def forward(self, x):
        new_features = super().forward(x)
        if self.drop_rate > 0:
            new_features = F.dropout(new_features, p=self.drop_rate, training=self.training)
        return torch.cat([x, new_features], 1)

    def layer_conv_info(self):
        return [1, 3], [1, 1], [0, 1]


class _DenseBlock(nn.Sequential):
    def __init__(self, num_layers, num_input_features, bn_size, growth_rate, drop_rate):
        super().__init__()
        self.block_kernel_sizes = []
        self.block_strides = []
        self.block_paddings = []
        for i in range(num_layers):
            layer = _DenseLayer(num_input_features + i * growth_rate,
                                growth_rate, bn_size, drop_rate)
            ks, ss, ps = layer.layer_conv_info()
            self.block_kernel_sizes.extend(ks)
            self.block_strides.extend(ss)
            self.block_paddings.extend(ps)
            self.add_module('denselayer%d' % (i + 1), layer)
        self.num_layers = _DenseLayer.num_layers * num_layers

    def block_conv_info(self):
        return self.block_kernel_sizes, self.block_strides, self.block_paddings


class _Transition(nn.Sequential):
    num_layers = 1

    def __init__(self, num_input_features, num_output_features):
        super().__init__()
        self.add_module('norm', FusedBatchNorm2d(num_input_features,
                                                 fused_relu=True))
        self.add_module('relu', nn.Identity())
        self.add_module('conv', GemmConv2d(num_input_features,
                                           num_output_features,
                                           kernel_size=1, stride=1,
                                           bias=False))
        self.add_module('pool', nn.AvgPool2d(kernel_size=2, stride=2))

    def block_conv_info(self):
        return [1, 2], [1, 2], [0, 0]


class DenseNetFeatures(nn.Module):
    """DenseNet-BC trunk, headless, initial max-pool removed (stride 16)."""

    def __init__(self, growth_rate=32, block_config=(6, 12, 24, 16),
                 num_init_features=64, bn_size=4, drop_rate=0):
        super().__init__()
        self.kernel_sizes = []
        self.strides = []
        self.paddings = []
        self.n_layers = 0

        self.features = nn.Sequential(OrderedDict([
            ('conv0', nn.Conv2d(3, num_init_features, kernel_size=7, stride=2,
                                padding=3, bias=False)),
            ('norm0', FusedBatchNorm2d(num_init_features, fused_relu=True)),
            ('relu0', nn.Identity()),
            # no pool0: stride stays 16 overall (reference densenet_features.py:116)
        ]))
        self.kernel_sizes.append(7)
        self.strides.append(2)
        self.paddings.append(3)

        num_features = num_init_features
        for i, num_layers in enumerate(block_config):
            block = _DenseBlock(num_layers=num_layers, num_input_features=num_features,
                                bn_size=bn_size, growth_rate=growth_rate,
                                drop_rate=drop_rate)
            self.n_layers += block.num_layers
            ks, ss, ps = block.block_conv_info()
            self.kernel_sizes.extend(ks)
            self.strides.extend(ss)
            self.paddings.extend(ps)
            self.features.add_module('denseblock%d' % (i + 1), block)
            num_features = num_features + num_layers * growth_rate
            if i != len(block_config) - 1:
                trans = _Transition(num_features, num_features // 2)
                self.n_layers += trans.num_layers
                ks, ss, ps = trans.block_conv_info()
                self.kernel_sizes.extend(ks)
                self.strides.extend(ss)
                self.paddings.extend(ps)
                self.features.add_module('transition%d' % (i + 1), trans)
                num_features = num_features // 2

        self.features.add_module('norm5', FusedBatchNorm2d(num_features,
                                                           fused_relu=True))
        self.features.add_module('final_relu', nn.Identity())
        self.out_channels = num_features

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight)
            elif isinstance(m, nn.BatchNorm2d):
                nn.init.constant_(m.weight, 1)
                nn.init.constant_(m.bias, 0)

    def forward(self, x):
        return self.features(x)

    def conv_info(self):
        return self.kernel_sizes, self.strides, self.paddings

    def num_layers(self):
        return self.n_layers

    def __repr__(self):
        return 'densenet{}_features'.format(self.num_layers() + 2)


_KEY_PATTERN = re.compile(
    r'^(.*denselayer\d+\.(?:norm|relu|conv))\.((?:[12])\.(?:weight|bias|running_mean|running_var))$')


def _fix_legacy_keys(sd):
    """torchvision's old 'norm.1' -> 'norm1' key fixup (reference :192-207)."""
    for key in list(sd.keys()):
        res = _KEY_PATTERN.match(key)
        if res:
            sd[res.group(1) + res.group(2)] = sd.pop(key)
    sd.pop('classifier.weight', None)
    sd.pop('classifier.bias', None)
    return sd


def _densenet(arch, growth_rate, block_config, num_init_features, pretrained, **kwargs):
    model = DenseNetFeatures(growth_rate=growth_rate, block_config=block_config,
                             num_init_features=num_init_features, **kwargs)
    if pretrained:
        path = os.path.join(PRETRAINED_DIR, f'{arch}.pth')
        if os.path.isfile(path):
            sd = torch.load(path, map_location='cpu', weights_only=False)
            model.load_state_dict(_fix_legacy_keys(sd), strict=False)
        else:
            warnings.warn(f'pretrained weights not found at {path}; using random init')
    return model


def densenet121_features(pretrained=False, **kwargs):
    return _densenet('densenet121', 32, (6, 12, 24, 16), 64, pretrained, **kwargs)


def densenet161_features(pretrained=False, **kwargs):
    return _densenet('densenet161', 48, (6, 12, 36, 24), 96, pretrained, **kwargs)


def densenet169_features(pretrained=False, **kwargs):
    return _densenet('densenet169', 32, (6, 12, 32, 32), 64, pretrained, **kwargs)


def densenet201_features(pretrained=False, **kwargs):
    return _densenet('densenet201', 32, (6, 12, 48, 32), 64, pretrained, **kwargs)
