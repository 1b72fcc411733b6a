"""paddle.nn.utils (reference: python/paddle/nn/utils/__init__.py)."""
from __future__ import annotations

import torch

weight_norm = torch.nn.utils.weight_norm
remove_weight_norm = torch.nn.utils.remove_weight_norm
spectral_norm = torch.nn.utils.spectral_norm


def parameters_to_vector(parameters, name=None):
    return torch.nn.utils.parameters_to_vector(list(parameters))


def vector_to_parameters(vec, parameters, name=None):
    return torch.nn.utils.vector_to_parameters(vec, list(parameters))


def clip_grad_norm_(parameters, max_norm, norm_type=2.0,
                    error_if_nonfinite=False):
    return torch.nn.utils.clip_grad_norm_(
        list(parameters), max_norm, norm_type,
        error_if_nonfinite=error_if_nonfinite)


def clip_grad_value_(parameters, clip_value):
    return torch.nn.utils.clip_grad_value_(list(parameters), clip_value)
