"""OpTest harness (reference: test/legacy_test/op_test.py:418 OpTest +
get_numeric_gradient:148): declare inputs + a callable, check outputs
against a numpy/torch oracle and analytic grads against central-difference
numeric grads with per-op thresholds."""
from __future__ import annotations

from typing import Callable, Dict, Sequence

import numpy as np
import torch


def numeric_gradient(fn: Callable, inputs: Sequence[torch.Tensor], idx: int,
                     delta: float = 1e-3) -> torch.Tensor:
    """Central-difference dL/dx_idx where L = sum(fn(*inputs))."""
    x = inputs[idx]
    grad = torch.zeros_like(x, dtype=torch.float64)
    flat = x.reshape(-1)
    gflat = grad.reshape(-1)
    for i in range(flat.numel()):
        orig = flat[i].item()
        flat[i] = orig + delta
        with torch.no_grad():
            hi = float(fn(*inputs).double().sum())
        flat[i] = orig - delta
        with torch.no_grad():
            lo = float(fn(*inputs).double().sum())
        flat[i] = orig
        gflat[i] = (hi - lo) / (2 * delta)
    return grad


class OpTest:
    """Subclass and set: self.fn (callable), self.inputs (list of tensors,
    float64 recommended), optional self.oracle (callable) for forward."""

    fn: Callable = None
    oracle: Callable = None
    rtol = 1e-4
    atol = 1e-5
    grad_rtol = 5e-3
    grad_atol = 1e-3

    def make_inputs(self):  # pragma: no cover - abstract
        raise NotImplementedError

    def check_output(self):
        inputs = self.make_inputs()
        out = self.fn(*inputs)
        ref = self.oracle(*inputs)
        torch.testing.assert_close(out.double(), ref.double(),
                                   rtol=self.rtol, atol=self.atol)

    def check_grad(self, input_indices=None):
        inputs = [t.detach().clone().double().requires_grad_(t.is_floating_point())
                  for t in self.make_inputs()]
        out = self.fn(*inputs)
        out.sum().backward()
        idxs = input_indices if input_indices is not None else [
            i for i, t in enumerate(inputs) if t.is_floating_point()]
        for i in idxs:
            num = numeric_gradient(self.fn, [t.detach().clone() for t in inputs], i)
            torch.testing.assert_close(inputs[i].grad, num,
                                       rtol=self.grad_rtol, atol=self.grad_atol)
