"""paddle.distribution parity (reference: python/paddle/distribution/ --
~25 distributions).  Thin adapters over torch.distributions with
paddle's method names (sample/rsample/log_prob/prob/entropy/kl_divergence).
"""
from __future__ import annotations

import torch
import torch.distributions as td


class Distribution:
    def __init__(self, t):
        self._t = t

    def sample(self, shape=()):
        return self._t.sample(torch.Size(shape))

    def rsample(self, shape=()):
        return self._t.rsample(torch.Size(shape))

    def log_prob(self, value):
        return self._t.log_prob(value)

    def prob(self, value):
        return self._t.log_prob(value).exp()

    def entropy(self):
        return self._t.entropy()

    @property
    def mean(self):
        return self._t.mean

    @property
    def variance(self):
        return self._t.variance

    def kl_divergence(self, other):
        return td.kl_divergence(self._t, other._t)


def _wrap(tcls):
    class _D(Distribution):
        def __init__(self, *args, **kwargs):
            args = [torch.as_tensor(a, dtype=torch.float32)
                    if isinstance(a, (int, float, list)) else a for a in args]
            kwargs = {k: (torch.as_tensor(v, dtype=torch.float32)
                          if isinstance(v, (int, float, list)) else v)
                      for k, v in kwargs.items()}
            super().__init__(tcls(*args, **kwargs))

    _D.__name__ = tcls.__name__
    return _D


class Normal(Distribution):
    def __init__(self, loc, scale, name=None):
        loc = torch.as_tensor(loc, dtype=torch.float32) if not isinstance(loc, torch.Tensor) else loc
        scale = torch.as_tensor(scale, dtype=torch.float32) if not isinstance(scale, torch.Tensor) else scale
        super().__init__(td.Normal(loc, scale))
        self.loc, self.scale = loc, scale


class Uniform(Distribution):
    def __init__(self, low, high, name=None):
        low = torch.as_tensor(low, dtype=torch.float32) if not isinstance(low, torch.Tensor) else low
        high = torch.as_tensor(high, dtype=torch.float32) if not isinstance(high, torch.Tensor) else high
        super().__init__(td.Uniform(low, high))
        self.low, self.high = low, high


class Categorical(Distribution):
    def __init__(self, logits=None, probs=None, name=None):
        if logits is not None and probs is None:
            super().__init__(td.Categorical(logits=torch.as_tensor(logits)))
        else:
            super().__init__(td.Categorical(probs=torch.as_tensor(probs if probs is not None else logits)))


Bernoulli = _wrap(td.Bernoulli)
Beta = _wrap(td.Beta)
Binomial = _wrap(td.Binomial)
Cauchy = _wrap(td.Cauchy)
Chi2 = _wrap(td.Chi2)
ContinuousBernoulli = _wrap(td.ContinuousBernoulli)
Dirichlet = _wrap(td.Dirichlet)
Exponential = _wrap(td.Exponential)
Gamma = _wrap(td.Gamma)
Geometric = _wrap(td.Geometric)
Gumbel = _wrap(td.Gumbel)
Laplace = _wrap(td.Laplace)
LogNormal = _wrap(td.LogNormal)
Multinomial = _wrap(td.Multinomial)
MultivariateNormal = _wrap(td.MultivariateNormal)
Poisson = _wrap(td.Poisson)
StudentT = _wrap(td.StudentT)


def kl_divergence(p, q):
    return td.kl_divergence(p._t if isinstance(p, Distribution) else p,
                            q._t if isinstance(q, Distribution) else q)


# long-tail (reference: distribution/__init__.py)
import torch as _t

ExponentialFamily = _t.distributions.ExponentialFamily
Independent = _t.distributions.Independent
class TransformedDistribution(_t.distributions.TransformedDistribution):
    """Accepts paddle-style Transform wrappers (unwraps to torch) or raw
    torch transforms (reference transformed_distribution.py)."""

    def __init__(self, base, transforms):
        base_t = base._t if isinstance(base, Distribution) else base
        if not isinstance(transforms, (list, tuple)):
            transforms = [transforms]
        tts = [getattr(t, "_t", t) for t in transforms]
        super().__init__(base_t, tts)
LKJCholesky = _t.distributions.LKJCholesky
register_kl = _t.distributions.register_kl


from .transform import (  # noqa: F401,E402
    AbsTransform, AffineTransform, ChainTransform, ExpTransform,
    IndependentTransform, PowerTransform, ReshapeTransform,
    SigmoidTransform, SoftmaxTransform, StackTransform,
    StickBreakingTransform, TanhTransform, Transform)
from . import transform  # noqa: F401,E402
