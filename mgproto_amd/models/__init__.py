"""Backbone feature extractors (PyTorch-ROCm / MIOpen path).

Per the MI355X-first design, standard conv stacks run through MIOpen —
hand-written HIP kernels are reserved for the prototype math in
``mgproto_amd.ops``.
"""

from .resnet import (resnet18_features, resnet34_features, resnet50_features,
                     resnet101_features, resnet152_features)
from .densenet import (densenet121_features, densenet161_features,
                       densenet169_features, densenet201_features)
from .vgg import (vgg11_features, vgg11_bn_features, vgg13_features,
                  vgg13_bn_features, vgg16_features, vgg16_bn_features,
                  vgg19_features, vgg19_bn_features)

__all__ = [
    'resnet18_features', 'resnet34_features', 'resnet50_features',
    'resnet101_features', 'resnet152_features',
    'densenet121_features', 'densenet161_features', 'densenet169_features',
    'densenet201_features',
    'vgg11_features', 'vgg11_bn_features', 'vgg13_features', 'vgg13_bn_features',
    'vgg16_features', 'vgg16_bn_features', 'vgg19_features', 'vgg19_bn_features',
]
