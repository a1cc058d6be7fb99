"""Analytic receptive-field propagation through a conv stack.

Same math as the reference (``/root/reference/utils/receptive_field.py``):
the standard (n, jump, rf-size, center) recurrence of Dang Ha The Hien's
receptive-field arithmetic, used to map a prototype's latent (h, w) index to
an input-pixel bounding box.
"""

import math
from typing import List, Sequence


def compute_layer_rf_info(layer_filter_size, layer_stride, layer_padding,
                          previous_layer_rf_info):
    n_in, j_in, r_in, start_in = previous_layer_rf_info

    if layer_padding == 'SAME':
        n_out = math.ceil(float(n_in) / float(layer_stride))
        if n_in % layer_stride == 0:
            pad = max(layer_filter_size - layer_stride, 0)
        else:
            pad = max(layer_filter_size - (n_in % layer_stride), 0)
    elif layer_padding == 'VALID':
        n_out = math.ceil(float(n_in - layer_filter_size + 1) / float(layer_stride))
        pad = 0
    else:
        pad = layer_padding * 2
        n_out = math.floor((n_in - layer_filter_size + pad) / layer_stride) + 1

    assert n_out == math.floor((n_in - layer_filter_size + pad) / layer_stride) + 1
    pL = math.floor(pad / 2)

    j_out = j_in * layer_stride
    r_out = r_in + (layer_filter_size - 1) * j_in
    start_out = start_in + ((layer_filter_size - 1) / 2 - pL) * j_in
    return [n_out, j_out, r_out, start_out]


def compute_proto_layer_rf_info_v2(img_size: int,
                                   layer_filter_sizes: Sequence[int],
                                   layer_strides: Sequence[int],
                                   layer_paddings: Sequence[int],
                                   prototype_kernel_size: int) -> List:
    assert len(layer_filter_sizes) == len(layer_strides) == len(layer_paddings)
    rf_info = [img_size, 1, 1, 0.5]
    for f, s, p in zip(layer_filter_sizes, layer_strides, layer_paddings):
        rf_info = compute_layer_rf_info(f, s, p, rf_info)
    return compute_layer_rf_info(prototype_kernel_size, 1, 'VALID', rf_info)


def compute_rf_protoL_at_spatial_location(img_size, height_index, width_index,
                                          protoL_rf_info):
    n, j, r, start = protoL_rf_info
    assert height_index < n and width_index < n
    center_h = start + height_index * j
    center_w = start + width_index * j
    return [max(int(center_h - r / 2), 0), min(int(center_h + r / 2), img_size),
            max(int(center_w - r / 2), 0), min(int(center_w + r / 2), img_size)]


def compute_rf_prototype(img_size, prototype_patch_index, protoL_rf_info):
    img_index, height_index, width_index = prototype_patch_index
    rf = compute_rf_protoL_at_spatial_location(img_size, height_index,
                                               width_index, protoL_rf_info)
    return [img_index, rf[0], rf[1], rf[2], rf[3]]


def compute_rf_prototypes(img_size, prototype_patch_indices, protoL_rf_info):
    return [compute_rf_prototype(img_size, p, protoL_rf_info)
            for p in prototype_patch_indices]
