"""paddle.incubate.autograd (reference: incubate/autograd/__init__.py --
functional AD: vjp/jvp/Jacobian/Hessian + prim toggles)."""
from __future__ import annotations

import torch

from ..autograd import jvp, vjp  # noqa: F401


class Jacobian:
    """Lazy full Jacobian of func at xs (reference: autograd/functional.py
    Jacobian)."""

    def __init__(self, func, xs, is_batched=False):
        self._J = torch.autograd.functional.jacobian(
            func, xs if isinstance(xs, tuple) else (xs,))
        if isinstance(self._J, tuple) and len(self._J) == 1:
            self._J = self._J[0]

    def __getitem__(self, idx):
        return self._J[idx]

    @property
    def shape(self):
        return self._J.shape


class Hessian:
    def __init__(self, func, xs, is_batched=False):
        self._H = torch.autograd.functional.hessian(
            func, xs if isinstance(xs, tuple) else xs)

    def __getitem__(self, idx):
        return self._H[idx]

    @property
    def shape(self):
        return self._H.shape


def forward_grad(outputs, inputs, grad_inputs=None):
    """Forward-mode grad (jvp over the computation)."""
    raise NotImplementedError(
        "forward_grad over recorded graphs targets the prim system; use "
        "paddle.incubate.autograd.jvp on a callable instead")


def grad(outputs, inputs, grad_outputs=None):
    out = torch.autograd.grad(outputs, inputs, grad_outputs,
                              allow_unused=True, retain_graph=True)
    return out[0] if len(out) == 1 else out


def enable_prim():
    """The prim (composite-op) system is replaced by eager composition on
    this stack; the toggle is accepted for compatibility."""


def disable_prim():
    pass
