"""Headless VGG feature extractors.

Re-implementation of the reference backbone contract
(``/root/reference/models/vgg_features.py``): VGG-11/13/16/19 (+BN variants)
trunks with the classifier removed, options to drop the final max-pool
(default: dropped, so output stride is 16) and the final ReLU
(reference vgg_features.py:58-94), per-layer ``conv_info()``
(vgg_features.py:42-56), and ``VGG_vanilla`` (vgg_features.py:110-124).
Module naming matches torchvision (``features.N``) so reference
checkpoints load directly. The norm->relu pairs run as
FusedBatchNorm2d(fused_relu=True) on GPU (models/fused_bn.py).
"""

import os
import warnings

import torch
import torch.nn as nn

from .resnet import PRETRAINED_DIR
from .fused_bn import FusedBatchNorm2d

cfg = {
    'A': [64, 'M', 128, 'M', 256, 256, 'M', 512, 512, 'M', 512, 512, 'M'],
    'B': [64, 64, 'M', 128, 128, 'M', 256, 256, 'M', 512, 512, 'M', 512, 512, 'M'],
    'D': [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 'M', 512, 512, 512, 'M',
          512, 512, 512, 'M'],
    'E': [64, 64, 'M', 128, 128, 'M', 256, 256, 256, 256, 'M', 512, 512, 512, 512,
          'M', 512, 512, 512, 512, 'M'],
}


class VGGFeatures(nn.Module):
    def __init__(self, layer_cfg, batch_norm=False, init_weights=True,
                 final_maxpool=False, final_relu=True):
        super().__init__()
        self.batch_norm = batch_norm
        self.kernel_sizes = []
        self.strides = []
        self.paddings = []
        self.features = self._make_layers(layer_cfg, batch_norm, final_maxpool,
                                          final_relu)
        self.out_channels = 512
        if init_weights:
            self._initialize_weights()

    def _make_layers(self, layer_cfg, batch_norm, final_maxpool, final_relu):
        self.n_layers = 0
        layers = []
        in_channels = 3
        for i, v in enumerate(layer_cfg):
            if v == 'M':
                if i == len(layer_cfg) - 1 and not final_maxpool:
                    continue  # drop the final max-pool: output stride 16
                layers += [nn.MaxPool2d(kernel_size=2, stride=2)]
                self.kernel_sizes.append(2)
                self.strides.append(2)
                self.paddings.append(0)
            else:
                conv2d = nn.Conv2d(in_channels, v, kernel_size=3, padding=1)
                if batch_norm:
                    layers += [conv2d, FusedBatchNorm2d(v, fused_relu=True),
                               nn.Identity()]
                elif i >= len(layer_cfg) - 2 and not final_relu:
                    layers += [conv2d]
                else:
                    layers += [conv2d, nn.ReLU(inplace=True)]
                self.n_layers += 1
                self.kernel_sizes.append(3)
                self.strides.append(1)
                self.paddings.append(1)
                in_channels = v
        return nn.Sequential(*layers)

    def _initialize_weights(self):
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode='fan_out', nonlinearity='relu')
                if m.bias is not None:
                    nn.init.constant_(m.bias, 0)
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
        return 'VGG{}, batch_norm={}'.format(self.n_layers + 3, self.batch_norm)


def _vgg(arch, layer_cfg, batch_norm, pretrained, **kwargs):
    model = VGGFeatures(cfg[layer_cfg], batch_norm=batch_norm, **kwargs)
    if pretrained:
        path = os.path.join(PRETRAINED_DIR, f'{arch}.pth')
        if os.path.isfile(path):
            sd = torch.load(path, map_location='cpu', weights_only=False)
            for k in list(sd.keys()):
                if k.startswith('classifier.'):
                    sd.pop(k)
            model.load_state_dict(sd, strict=False)
        else:
            warnings.warn(f'pretrained weights not found at {path}; using random init')
    return model


def vgg11_features(pretrained=False, **kwargs):
    return _vgg('vgg11', 'A', False, pretrained, **kwargs)


def vgg11_bn_features(pretrained=False, **kwargs):
    return _vgg('vgg11_bn', 'A', True, pretrained, **kwargs)


def vgg13_features(pretrained=False, **kwargs):
    return _vgg('vgg13', 'B', False, pretrained, **kwargs)


def vgg13_bn_features(pretrained=False, **kwargs):
    return _vgg('vgg13_bn', 'B', True, pretrained, **kwargs)


def vgg16_features(pretrained=False, **kwargs):
    return _vgg('vgg16', 'D', False, pretrained, **kwargs)


def vgg16_bn_features(pretrained=False, **kwargs):
    return _vgg('vgg16_bn', 'D', True, pretrained, **kwargs)


def vgg19_features(pretrained=False, **kwargs):
    return _vgg('vgg19', 'E', False, pretrained, **kwargs)


def vgg19_bn_features(pretrained=False, **kwargs):
    return _vgg('vgg19_bn', 'E', True, pretrained, **kwargs)


class VGG_vanilla(nn.Module):
    """Plain VGG-19 classifier head over the full trunk (reference
    vgg_features.py:110-124); used for baseline comparisons only."""

    def __init__(self, num_classes=200, pretrained=False):
        super().__init__()
        self.vgg19_f = vgg19_features(pretrained=pretrained,
                                      final_maxpool=True, final_relu=True)
        self.addons = nn.Linear(512 * 7 * 7, num_classes)

    def forward(self, x):
        x = self.vgg19_f(x)
        return self.addons(x.flatten(1))
