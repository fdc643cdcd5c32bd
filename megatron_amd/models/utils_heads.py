"""Small head helpers (reference megatron/model/utils.py)."""

import torch


def get_linear_layer(rows, columns, init_method):
    layer = torch.nn.Linear(rows, columns)
    init_method(layer.weight)
    with torch.no_grad():
        layer.bias.zero_()
    return layer
