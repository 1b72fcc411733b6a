"""paddle.optimizer parity (reference: python/paddle/optimizer/optimizer.py:127
Optimizer base; adamw.py:49; adam.py:58; momentum/sgd).

MI355X design: AdamW/Adam keep fp32 master weights for bf16/fp16 params
(multi_precision default on for low-precision params) and update through
the fused gfx950 kernel (paddle_amd.ops.fused_adamw_step) -- one kernel
launch per parameter, fp32 math, bf16 weight write-back.  The
flat-shard variants used by Fleet sharding live in
paddle_amd.distributed.fleet.sharding.
"""
from __future__ import annotations

import math
from typing import Iterable, Optional

import torch

from .. import framework
from ..ops import functional as hot
from . import lr as lr_mod
from .lr import LRScheduler


class Optimizer:
    def __init__(self, learning_rate=0.001, parameters=None, weight_decay=None,
                 grad_clip=None, name=None, multi_precision=None):
        self._lr = learning_rate
        if parameters is None:
            # static-graph style: parameters bound later by minimize()
            from .. import static as _static
            if not _static.in_static_mode():
                raise ValueError("parameters must be given in dygraph mode")
            parameters = []
        self._param_groups = list(parameters)
        if self._param_groups and isinstance(self._param_groups[0], dict):
            self._params = [p for g in self._param_groups for p in g["params"]]
        else:
            self._params = list(self._param_groups)
        self._weight_decay = 0.0 if weight_decay is None else (
            weight_decay if isinstance(weight_decay, float) else float(weight_decay))
        self._grad_clip = grad_clip
        self._multi_precision = multi_precision
        self._accumulators = {}  # name -> {id(param): tensor}
        self._masters = {}
        self._step_count = 0

    # -- lr -----------------------------------------------------------------
    def get_lr(self):
        if isinstance(self._lr, LRScheduler):
            return self._lr()
        return self._lr

    def set_lr(self, value):
        self._lr = value

    def _get_master(self, p):
        if p.dtype in (torch.bfloat16, torch.float16) and self._use_multi_precision():
            key = id(p)
            if key not in self._masters:
                self._masters[key] = p.detach().float().clone()
            return self._masters[key]
        return None

    def _use_multi_precision(self):
        return self._multi_precision is not False

    def _acc(self, name, p, init=None):
        d = self._accumulators.setdefault(name, {})
        key = id(p)
        if key not in d:
            master = self._get_master(p)
            ref = master if master is not None else p
            d[key] = torch.zeros_like(ref, dtype=torch.float32) if init is None else init(ref)
        return d[key]

    # -- grads ----------------------------------------------------------------
    def clear_grad(self, set_to_zero=True):
        for p in self._params:
            if p.grad is not None:
                if set_to_zero:
                    p.grad.zero_()
                else:
                    p.grad = None

    clear_gradients = clear_grad

    def _clip_grads(self):
        clip = self._grad_clip
        if clip is None:
            return 1.0
        from ..nn import ClipGradByGlobalNorm, ClipGradByNorm, ClipGradByValue
        if isinstance(clip, ClipGradByGlobalNorm):
            total = None
            for p in self._params:
                if p.grad is None:
                    continue
                sq = hot.l2_norm_squared(p.grad)
                total = sq if total is None else total + sq
            if total is None:
                return 1.0
            gnorm = total.sqrt()
            scale = clip.clip_norm / torch.clamp(gnorm, min=clip.clip_norm)
            for p in self._params:
                if p.grad is not None:
                    p.grad.mul_(scale.to(p.grad.dtype))
            return scale
        if isinstance(clip, ClipGradByNorm):
            for p in self._params:
                if p.grad is None:
                    continue
                n = p.grad.float().norm()
                if n > clip.clip_norm:
                    p.grad.mul_((clip.clip_norm / n).to(p.grad.dtype))
            return 1.0
        if isinstance(clip, ClipGradByValue):
            for p in self._params:
                if p.grad is not None:
                    p.grad.clamp_(clip.min, clip.max)
            return 1.0
        return 1.0

    @torch.no_grad()
    def step(self):
        self._clip_grads()
        self._step_count += 1
        for p in self._params:
            if p.grad is None or not p.requires_grad:
                continue
            self._apply_one(p)

    def _apply_one(self, p):  # pragma: no cover - abstract
        raise NotImplementedError

    # -- static-graph entry points (reference: optimizer.minimize) -----------
    def minimize(self, loss, startup_program=None, parameters=None,
                 no_grad_set=None):
        from .. import static as _static
        if isinstance(loss, _static.Var):
            _static.default_main_program().train_ops.append((loss, self))
            return None, None
        loss.backward()
        self.step()
        self.clear_grad()
        return None, None

    def _static_step(self, prog):
        """One optimizer application over the program's parameters."""
        if not self._params:
            self._params = [v.tensor for v in prog.params]
        self.step()
        self.clear_grad()

    # -- state dict (feeds .pdopt format) -------------------------------------
    def state_dict(self):
        sd = {}
        for name, d in self._accumulators.items():
            for i, p in enumerate(self._params):
                if id(p) in d:
                    sd[f"{_pname(p, i)}_{name}"] = d[id(p)]
        for i, p in enumerate(self._params):
            if id(p) in self._masters:
                sd[f"{_pname(p, i)}_master"] = self._masters[id(p)]
        if isinstance(self._lr, LRScheduler):
            sd["LR_Scheduler"] = self._lr.state_dict()
        sd["global_step"] = self._step_count
        return sd

    def set_state_dict(self, state):
        import numpy as np
        self._step_count = int(state.get("global_step", 0))
        if "LR_Scheduler" in state and isinstance(self._lr, LRScheduler):
            self._lr.set_state_dict(state["LR_Scheduler"])
        for name in list(self._accumulators.keys()) or ["moment1", "moment2"]:
            pass
        for i, p in enumerate(self._params):
            pn = _pname(p, i)
            for name in ("moment1", "moment2", "beta1_pow_acc", "beta2_pow_acc", "velocity"):
                k = f"{pn}_{name}"
                if k in state:
                    v = state[k]
                    if isinstance(v, np.ndarray):
                        v = torch.from_numpy(v)
                    self._accumulators.setdefault(name, {})[id(p)] = v.to(p.device).float()
            mk = f"{pn}_master"
            if mk in state:
                v = state[mk]
                if isinstance(v, np.ndarray):
                    v = torch.from_numpy(v)
                self._masters[id(p)] = v.to(p.device).float()

    set_dict = set_state_dict


def _pname(p, i):
    n = getattr(p, "name", None)
    return n if isinstance(n, str) else f"param_{i}"


class AdamW(Optimizer):
    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999, epsilon=1e-8,
                 parameters=None, weight_decay=0.01, lr_ratio=None, apply_decay_param_fun=None,
                 grad_clip=None, lazy_mode=False, multi_precision=None, name=None):
        super().__init__(learning_rate, parameters, weight_decay, grad_clip, name,
                         multi_precision)
        self._beta1, self._beta2, self._eps = beta1, beta2, epsilon
        self._apply_decay_param_fun = apply_decay_param_fun

    def _apply_one(self, p):
        lr = self.get_lr()
        wd = self._weight_decay
        if self._apply_decay_param_fun is not None and not self._apply_decay_param_fun(_pname(p, 0)):
            wd = 0.0
        master = self._get_master(p)
        m = self._acc("moment1", p)
        v = self._acc("moment2", p)
        if master is not None:
            hot.fused_adamw_step(master.view(-1), p.data.view(-1), p.grad.view(-1),
                                 m.view(-1), v.view(-1), lr, self._beta1, self._beta2,
                                 self._eps, wd, self._step_count)
        else:
            tgt = p.data
            if tgt.dtype != torch.float32:
                f = tgt.float()
                hot.fused_adamw_step(f.view(-1), None, p.grad.view(-1), m.view(-1),
                                     v.view(-1), lr, self._beta1, self._beta2,
                                     self._eps, wd, self._step_count)
                tgt.copy_(f.to(tgt.dtype))
            else:
                hot.fused_adamw_step(tgt.view(-1), None, p.grad.view(-1), m.view(-1),
                                     v.view(-1), lr, self._beta1, self._beta2,
                                     self._eps, wd, self._step_count)


class Adam(AdamW):
    """paddle.optimizer.Adam -- L2 regularization folded into grad (here:
    weight_decay defaults to None == 0, matching paddle's default)."""

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999, epsilon=1e-8,
                 parameters=None, weight_decay=None, grad_clip=None, lazy_mode=False,
                 multi_precision=None, name=None):
        super().__init__(learning_rate, beta1, beta2, epsilon, parameters,
                         0.0 if weight_decay is None else weight_decay, None, None,
                         grad_clip, lazy_mode, multi_precision, name)

    # Adam applies L2 as grad += wd * p (not decoupled); paddle's default
    # regularizer is None so the common case matches AdamW with wd=0.
    def _apply_one(self, p):
        if self._weight_decay:
            p.grad.add_(p.data.to(p.grad.dtype), alpha=self._weight_decay)
            wd_save = self._weight_decay
            self._weight_decay = 0.0
            super()._apply_one(p)
            self._weight_decay = wd_save
        else:
            super()._apply_one(p)


class Momentum(Optimizer):
    def __init__(self, learning_rate=0.001, momentum=0.9, parameters=None,
                 use_nesterov=False, weight_decay=None, grad_clip=None,
                 multi_precision=False, name=None):
        super().__init__(learning_rate, parameters, weight_decay, grad_clip, name,
                         multi_precision)
        self._momentum = momentum
        self._nesterov = use_nesterov

    def _apply_one(self, p):
        lr = self.get_lr()
        vel = self._acc("velocity", p)
        g = p.grad.float()
        if self._weight_decay:
            g = g + self._weight_decay * p.data.float()
        vel.mul_(self._momentum).add_(g)
        if self._nesterov:
            upd = g + self._momentum * vel
        else:
            upd = vel
        p.data.add_(-lr * upd.to(p.dtype))


class SGD(Optimizer):
    def __init__(self, learning_rate=0.001, parameters=None, weight_decay=None,
                 grad_clip=None, multi_precision=False, name=None):
        super().__init__(learning_rate, parameters, weight_decay, grad_clip, name,
                         multi_precision)

    def _apply_one(self, p):
        lr = self.get_lr()
        g = p.grad
        if self._weight_decay:
            g = g + self._weight_decay * p.data
        p.data.add_(g, alpha=-lr)


class Lamb(Optimizer):
    def __init__(self, learning_rate=0.001, lamb_weight_decay=0.01, beta1=0.9,
                 beta2=0.999, epsilon=1e-6, parameters=None, grad_clip=None,
                 exclude_from_weight_decay_fn=None, multi_precision=False, name=None):
        super().__init__(learning_rate, parameters, lamb_weight_decay, grad_clip,
                         name, multi_precision)
        self._beta1, self._beta2, self._eps = beta1, beta2, epsilon
        self._exclude = exclude_from_weight_decay_fn

    def _apply_one(self, p):
        lr = self.get_lr()
        m = self._acc("moment1", p)
        v = self._acc("moment2", p)
        g = p.grad.float()
        t = self._step_count
        m.mul_(self._beta1).add_(g, alpha=1 - self._beta1)
        v.mul_(self._beta2).addcmul_(g, g, value=1 - self._beta2)
        mhat = m / (1 - self._beta1 ** t)
        vhat = v / (1 - self._beta2 ** t)
        wd = 0.0 if (self._exclude and self._exclude(p)) else self._weight_decay
        r = mhat / (vhat.sqrt() + self._eps) + wd * p.data.float()
        w_norm = p.data.float().norm()
        r_norm = r.norm()
        ratio = torch.where((w_norm > 0) & (r_norm > 0), w_norm / r_norm,
                            torch.ones_like(w_norm))
        p.data.add_((-lr * ratio * r).to(p.dtype))


lr = lr_mod


# ---------------------------------------------------------------------------
# long-tail optimizers (reference: optimizer/{adagrad,adamax,asgd,radam,
# rmsprop,adadelta,rprop,nadam,lbfgs}.py) -- torch.optim engines under the
# paddle Optimizer API (learning_rate/parameters/clear_grad/.pdopt dicts)
# ---------------------------------------------------------------------------
class _TorchOptimizer(Optimizer):
    _torch_cls = None
    _kw_map = {}

    def __init__(self, learning_rate=0.001, parameters=None, weight_decay=None,
                 grad_clip=None, name=None, multi_precision=None, **kwargs):
        super().__init__(learning_rate=learning_rate, parameters=parameters,
                         weight_decay=weight_decay, grad_clip=grad_clip)
        tkw = {}
        for pk, tk in self._kw_map.items():
            if pk in kwargs and kwargs[pk] is not None:
                tkw[tk] = kwargs[pk]
        wd = self._weight_decay
        self._tkw = dict(tkw, weight_decay=wd)
        self._opt = None

    def _ensure(self):
        if self._opt is None and self._params:
            lr = self.get_lr()
            self._opt = self._torch_cls(self._params, lr=lr, **self._tkw)
        return self._opt

    @torch.no_grad()
    def step(self):
        self._clip_grads()
        self._step_count += 1
        opt = self._ensure()
        if opt is None:
            return
        for g in opt.param_groups:
            g["lr"] = self.get_lr()
        opt.step()

    def state_dict(self):
        opt = self._ensure()
        return opt.state_dict() if opt else {}

    def set_state_dict(self, sd):
        opt = self._ensure()
        if opt and sd:
            import copy
            # torch keeps (not copies) state tensors when dtype/device match;
            # without the deepcopy two optimizers would share mutable state
            opt.load_state_dict(copy.deepcopy(sd))


class Adagrad(_TorchOptimizer):
    _torch_cls = torch.optim.Adagrad
    _kw_map = {"epsilon": "eps", "initial_accumulator_value":
               "initial_accumulator_value"}

    def __init__(self, learning_rate=0.001, epsilon=1e-6,
                 initial_accumulator_value=0.0, **kw):
        super().__init__(learning_rate=learning_rate, epsilon=epsilon,
                         initial_accumulator_value=initial_accumulator_value,
                         **kw)


class Adamax(_TorchOptimizer):
    _torch_cls = torch.optim.Adamax

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-8, **kw):
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.update(betas=(beta1, beta2), eps=epsilon)


class ASGD(_TorchOptimizer):
    _torch_cls = torch.optim.ASGD


class RAdam(_TorchOptimizer):
    _torch_cls = torch.optim.RAdam

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-8, **kw):
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.update(betas=(beta1, beta2), eps=epsilon)


class NAdam(_TorchOptimizer):
    _torch_cls = torch.optim.NAdam

    def __init__(self, learning_rate=0.001, beta1=0.9, beta2=0.999,
                 epsilon=1e-8, momentum_decay=0.004, **kw):
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.update(betas=(beta1, beta2), eps=epsilon,
                         momentum_decay=momentum_decay)


class RMSProp(_TorchOptimizer):
    _torch_cls = torch.optim.RMSprop

    def __init__(self, learning_rate=0.001, rho=0.95, epsilon=1e-6,
                 momentum=0.0, centered=False, **kw):
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.update(alpha=rho, eps=epsilon, momentum=momentum,
                         centered=centered)


class Adadelta(_TorchOptimizer):
    _torch_cls = torch.optim.Adadelta

    def __init__(self, learning_rate=0.001, epsilon=1e-6, rho=0.95, **kw):
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.update(eps=epsilon, rho=rho)


class Rprop(_TorchOptimizer):
    _torch_cls = torch.optim.Rprop

    def __init__(self, learning_rate=0.001, learning_rate_range=(1e-5, 50),
                 etas=(0.5, 1.2), **kw):
        kw.pop("weight_decay", None)
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.pop("weight_decay", None)
        self._tkw.update(etas=tuple(etas), step_sizes=tuple(learning_rate_range))


class LBFGS(_TorchOptimizer):
    _torch_cls = torch.optim.LBFGS

    def __init__(self, learning_rate=1.0, max_iter=20, max_eval=None,
                 tolerance_grad=1e-7, tolerance_change=1e-9, history_size=100,
                 line_search_fn=None, **kw):
        kw.pop("weight_decay", None)
        super().__init__(learning_rate=learning_rate, **kw)
        self._tkw.pop("weight_decay", None)
        self._tkw.update(max_iter=max_iter, max_eval=max_eval,
                         tolerance_grad=tolerance_grad,
                         tolerance_change=tolerance_change,
                         history_size=history_size,
                         line_search_fn=line_search_fn)

    @torch.no_grad()
    def step(self, closure=None):
        opt = self._ensure()
        if closure is not None:
            with torch.enable_grad():
                return opt.step(closure)
        return opt.step(lambda: 0.0)
