"""Input normalization (reference ``utils/preprocess.py``)."""

import torch

mean = (0.485, 0.456, 0.406)
std = (0.229, 0.224, 0.225)


def preprocess(x: torch.Tensor, mean, std) -> torch.Tensor:
    assert x.size(1) == 3
    y = torch.zeros_like(x)
    for i in range(3):
        y[:, i] = (x[:, i] - mean[i]) / std[i]
    return y


def preprocess_input_function(x: torch.Tensor) -> torch.Tensor:
    """Normalize [0,1] images with ImageNet mean/std (used at push time —
    the push loader yields unnormalized images, reference push.py:14)."""
    return preprocess(x, mean=mean, std=std)


def undo_preprocess(x: torch.Tensor, mean, std) -> torch.Tensor:
    assert x.size(1) == 3
    y = torch.zeros_like(x)
    for i in range(3):
        y[:, i] = x[:, i] * std[i] + mean[i]
    return y


def undo_preprocess_input_function(x: torch.Tensor) -> torch.Tensor:
    return undo_preprocess(x, mean=mean, std=std)
