"""incubate optimizers (reference: incubate/optimizer/{lookahead,
modelaverage}.py)."""
from __future__ import annotations

import torch


class LookAhead:
    """Wraps an optimizer: every k steps, slow weights interpolate toward
    fast weights (reference: lookahead.py:LookAhead)."""

    def __init__(self, inner_optimizer, alpha=0.5, k=5, name=None):
        self.inner = inner_optimizer
        self.alpha = alpha
        self.k = k
        self._steps = 0
        self._slow = {}

    def step(self):
        self.inner.step()
        self._steps += 1
        if self._steps % self.k == 0:
            for p in self.inner._params:
                key = id(p)
                if key not in self._slow:
                    self._slow[key] = p.detach().clone()
                slow = self._slow[key]
                with torch.no_grad():
                    slow.add_(self.alpha * (p.detach() - slow))
                    p.copy_(slow)

    def clear_grad(self, *a, **kw):
        self.inner.clear_grad(*a, **kw)

    def minimize(self, loss, **kw):
        loss.backward()
        self.step()
        self.clear_grad()

    def __getattr__(self, name):
        return getattr(self.inner, name)


class ModelAverage:
    """Maintains an average of parameters over steps; apply() swaps the
    average in (reference: modelaverage.py:ModelAverage)."""

    def __init__(self, average_window_rate=0.15, parameters=None,
                 min_average_window=10000, max_average_window=10000, name=None):
        self._params = list(parameters or [])
        self._sum = {id(p): torch.zeros_like(p) for p in self._params}
        self._cnt = 0
        self._saved = {}

    def step(self):
        with torch.no_grad():
            for p in self._params:
                self._sum[id(p)].add_(p.detach())
        self._cnt += 1

    def apply(self, executor=None, need_restore=True):
        import contextlib

        @contextlib.contextmanager
        def ctx():
            with torch.no_grad():
                for p in self._params:
                    self._saved[id(p)] = p.detach().clone()
                    if self._cnt:
                        p.copy_(self._sum[id(p)] / self._cnt)
            try:
                yield
            finally:
                if need_restore:
                    self.restore()
        return ctx()

    def restore(self, executor=None):
        with torch.no_grad():
            for p in self._params:
                if id(p) in self._saved:
                    p.copy_(self._saved[id(p)])

    def minimize(self, loss, **kw):
        raise NotImplementedError("ModelAverage wraps evaluation, not training")


from ..optimizer import LBFGS  # noqa: E402,F401
