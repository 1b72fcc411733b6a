"""paddle.autograd parity (reference: python/paddle/autograd/).

backward/grad delegate to torch autograd; PyLayer maps onto
torch.autograd.Function with the paddle ctx API.
"""
from __future__ import annotations

import torch


def backward(tensors, grad_tensors=None, retain_graph=False):
    if not isinstance(tensors, (list, tuple)):
        tensors = [tensors]
    torch.autograd.backward(tensors, grad_tensors, retain_graph=retain_graph)


def grad(outputs, inputs, grad_outputs=None, retain_graph=None, create_graph=False,
         only_inputs=True, allow_unused=False, no_grad_vars=None):
    outs = outputs if isinstance(outputs, (list, tuple)) else [outputs]
    ins = inputs if isinstance(inputs, (list, tuple)) else [inputs]
    res = torch.autograd.grad(outs, ins, grad_outputs, retain_graph=retain_graph,
                              create_graph=create_graph, allow_unused=allow_unused)
    return list(res)


no_grad = torch.no_grad
enable_grad = torch.enable_grad
set_grad_enabled = torch.set_grad_enabled


def is_grad_enabled():
    return torch.is_grad_enabled()


class PyLayerContext:
    """ctx handed to PyLayer.forward/backward (reference:
    paddle/fluid/eager/pylayer/)."""

    def __init__(self, torch_ctx):
        self._ctx = torch_ctx

    def save_for_backward(self, *tensors):
        self._ctx.save_for_backward(*tensors)

    def saved_tensor(self):
        return self._ctx.saved_tensors

    def mark_not_inplace(self, *args):
        pass

    def mark_non_differentiable(self, *args):
        self._ctx.mark_non_differentiable(*args)

    def set_materialize_grads(self, v):
        self._ctx.set_materialize_grads(v)

    def __getattr__(self, k):
        return getattr(self.__dict__["_ctx"], k)

    def __setattr__(self, k, v):
        if k == "_ctx":
            object.__setattr__(self, k, v)
        else:
            setattr(self._ctx, k, v)


class PyLayerMeta(type):
    def __new__(mcls, name, bases, ns):
        cls = super().__new__(mcls, name, bases, ns)
        if name == "PyLayer":
            return cls
        fwd = ns.get("forward") or cls.forward
        bwd = ns.get("backward") or cls.backward

        class _Fn(torch.autograd.Function):
            @staticmethod
            def forward(tctx, *args, **kw):
                return fwd(PyLayerContext(tctx), *args, **kw)

            @staticmethod
            def backward(tctx, *grads):
                out = bwd(PyLayerContext(tctx), *grads)
                return out

        cls._torch_fn = _Fn
        return cls


class PyLayer(metaclass=PyLayerMeta):
    @classmethod
    def apply(cls, *args, **kwargs):
        return cls._torch_fn.apply(*args, **kwargs)

    @staticmethod
    def forward(ctx, *args, **kwargs):  # pragma: no cover - abstract
        raise NotImplementedError

    @staticmethod
    def backward(ctx, *grads):  # pragma: no cover - abstract
        raise NotImplementedError


# functional double-grad helpers
def vjp(func, xs, v=None):
    xs_l = xs if isinstance(xs, (list, tuple)) else [xs]
    for x in xs_l:
        x.requires_grad_(True)
    ys = func(*xs_l)
    ys_l = ys if isinstance(ys, (list, tuple)) else [ys]
    if v is None:
        v = [torch.ones_like(y) for y in ys_l]
    elif not isinstance(v, (list, tuple)):
        v = [v]
    grads = torch.autograd.grad(ys_l, xs_l, v, create_graph=True, allow_unused=True)
    return ys, list(grads)


def jvp(func, xs, v=None):
    xs_l = tuple(xs) if isinstance(xs, (list, tuple)) else (xs,)
    if v is None:
        v = tuple(torch.ones_like(x) for x in xs_l)
    elif not isinstance(v, (list, tuple)):
        v = (v,)
    ys, jv = torch.autograd.functional.jvp(func, xs_l, tuple(v), create_graph=True)
    return ys, list(jv) if isinstance(jv, tuple) else [jv]


def hessian(func, xs, batch_axis=None):
    return torch.autograd.functional.hessian(func, xs)


def jacobian(func, xs, batch_axis=None):
    return torch.autograd.functional.jacobian(func, xs)


# reference: autograd/saved_tensors_hooks.py -- pack/unpack hook context
from torch.autograd.graph import saved_tensors_hooks  # noqa: F401
