"""paddle.static: deferred-execution graph over the eager substrate.

Reference surface: python/paddle/static/ (Program, Executor, data,
default_main_program, nn.fc) and base/executor.py:1234 (Executor.run
with feed/fetch).

MI355X-first design: instead of rebuilding the reference's PIR program
translator + instruction scheduler, the static API records a DEFERRED
GRAPH of torch ops.  `paddle.static.data` returns a symbolic Var;
every torch function touching a Var (all paddle_amd ops dispatch to
torch functions) is intercepted via `__torch_function__` and recorded
as a graph node.  `Executor.run` binds feeds, evaluates the graph
(memoized per run) through the SAME HIP kernels as dygraph, and runs
the recorded optimizer step.  Semantics match the reference's
build-once / run-many static workflow; execution is eager per run,
which on MI355X is the right call: kernels are already fused and
hipGraph capture can wrap the run when launch-bound.
"""
from __future__ import annotations

import numpy as np
import torch

from .. import framework

_static_mode = False


def enable_static():
    global _static_mode
    _static_mode = True


def disable_static():
    global _static_mode
    _static_mode = False


def in_static_mode():
    return _static_mode


class Var:
    """Symbolic node in a static Program.

    kind: "data" (bound from feed), "param" (persistent tensor),
    "op" (fn over input Vars/constants).
    """

    def __init__(self, kind, name=None, shape=None, dtype=None, fn=None,
                 args=None, kwargs=None, tensor=None):
        self.kind = kind
        self.name = name
        self.shape = shape
        self.dtype = dtype
        self.fn = fn
        self.args = args or ()
        self.kwargs = kwargs or {}
        self.tensor = tensor          # params: the persistent torch tensor
        self.stop_gradient = kind != "param"

    # -- evaluation ---------------------------------------------------------
    def _eval(self, env, memo):
        key = id(self)
        if key in memo:
            return memo[key]
        if self.kind == "data":
            if self.name not in env:
                raise KeyError(f"feed missing for data var '{self.name}'")
            v = env[self.name]
        elif self.kind == "param":
            v = self.tensor
        else:
            args = [a._eval(env, memo) if isinstance(a, Var) else a
                    for a in self.args]
            kwargs = {k: (v._eval(env, memo) if isinstance(v, Var) else v)
                      for k, v in self.kwargs.items()}
            v = self.fn(*args, **kwargs)
        memo[key] = v
        return v

    # -- interception: any torch fn over a Var records an op node -----------
    @classmethod
    def __torch_function__(cls, func, types, args=(), kwargs=None):
        return Var("op", fn=func, args=args, kwargs=kwargs or {})

    # python operators route through torch so they are captured too
    def _binop(self, other, fn, swap=False):
        a, b = (other, self) if swap else (self, other)
        return Var("op", fn=fn, args=(a, b))

    def __add__(self, o):
        return self._binop(o, torch.add)

    __radd__ = __add__

    def __sub__(self, o):
        return self._binop(o, torch.subtract)

    def __rsub__(self, o):
        return self._binop(o, torch.subtract, swap=True)

    def __mul__(self, o):
        return self._binop(o, torch.multiply)

    __rmul__ = __mul__

    def __truediv__(self, o):
        return self._binop(o, torch.divide)

    def __rtruediv__(self, o):
        return self._binop(o, torch.divide, swap=True)

    def __matmul__(self, o):
        return self._binop(o, torch.matmul)

    def __neg__(self):
        return Var("op", fn=torch.neg, args=(self,))

    def __getitem__(self, idx):
        return Var("op", fn=lambda t, i: t[i], args=(self, idx))

    def reshape(self, shape):
        return Var("op", fn=torch.reshape, args=(self, shape))

    def astype(self, dtype):
        dt = framework.convert_dtype(dtype)
        return Var("op", fn=lambda t, d: t.to(d), args=(self, dt))

    def __repr__(self):
        return f"Var(kind={self.kind}, name={self.name}, shape={self.shape})"


class Program:
    """Recorded graph + optimizer step (reference: base/framework.py Program)."""

    def __init__(self):
        self.params = []              # persistent param Vars
        self.train_ops = []           # [(loss_var, optimizer)]
        self.initializers = []        # callables run by the startup program

    def clone(self, for_test=False):
        import copy
        p = copy.copy(self)
        if for_test:
            p = copy.copy(self)
            p.train_ops = []
        return p

    # reference API parity
    def global_block(self):
        return self

    def all_parameters(self):
        return list(self.params)


_default_main = Program()
_default_startup = Program()


def default_main_program():
    return _default_main


def default_startup_program():
    return _default_startup


class program_guard:
    def __init__(self, main_program, startup_program=None):
        self.main = main_program
        self.startup = startup_program

    def __enter__(self):
        global _default_main, _default_startup
        self._saved = (_default_main, _default_startup)
        _default_main = self.main
        if self.startup is not None:
            _default_startup = self.startup
        return self

    def __exit__(self, *a):
        global _default_main, _default_startup
        _default_main, _default_startup = self._saved


def data(name, shape, dtype="float32", lod_level=0):
    return Var("data", name=name, shape=shape,
               dtype=framework.convert_dtype(dtype))


def create_parameter(shape, dtype="float32", initializer=None, program=None):
    prog = program or _default_main
    dt = framework.convert_dtype(dtype)
    t = torch.empty(shape, dtype=dt)
    v = Var("param", shape=shape, dtype=dt, tensor=t)
    t.requires_grad_(True)

    def init():
        with torch.no_grad():
            if initializer is not None:
                initializer(t)
            else:
                torch.nn.init.xavier_normal_(t) if t.dim() >= 2 else t.zero_()
    prog.initializers.append(init)
    prog.params.append(v)
    return v


class nn:
    """paddle.static.nn subset (reference: python/paddle/static/nn/common.py)."""

    @staticmethod
    def fc(x, size, activation=None, name=None, num_flatten_dims=1,
           weight_attr=None, bias_attr=None):
        in_dim = x.shape[-1]
        if in_dim is None or in_dim < 0:
            raise ValueError("static.nn.fc needs a known last-dim size")
        w = create_parameter([in_dim, size])
        b = create_parameter([size], initializer=lambda t: t.zero_())
        out = Var("op", fn=torch.addmm, args=(b, x, w))
        if activation == "relu":
            out = Var("op", fn=torch.relu, args=(out,))
        elif activation == "softmax":
            out = Var("op", fn=lambda t: torch.softmax(t, -1), args=(out,))
        elif activation == "tanh":
            out = Var("op", fn=torch.tanh, args=(out,))
        out.shape = (list(x.shape[:-1]) + [size]) if x.shape else [None, size]
        return out


class Executor:
    """reference: python/paddle/base/executor.py:1234."""

    def __init__(self, place=None):
        self.place = place
        self._dev = (torch.device("cuda") if (place is not None and
                     "GPU" in type(place).__name__ and torch.cuda.is_available())
                     else torch.device("cpu"))

    def run(self, program=None, feed=None, fetch_list=None, return_numpy=True):
        prog = program if isinstance(program, Program) else _default_main
        # startup program: run pending initializers
        if prog is _default_startup or (isinstance(program, Program) and
                                        program.initializers and not program.train_ops
                                        and fetch_list is None and not feed):
            for init in (prog.initializers or _default_main.initializers):
                init()
            prog.initializers = []
            return []
        feed = feed or {}
        env = {}
        for k, v in feed.items():
            t = v if isinstance(v, torch.Tensor) else torch.as_tensor(np.asarray(v))
            env[k] = t.to(self._dev)
        memo = {}
        # legacy program captured via paddle.jit
        fn = getattr(prog, "fn", None)
        if fn is not None:
            outs = fn(**env)
            outs = outs if isinstance(outs, (list, tuple)) else [outs]
        else:
            for loss_var, opt in prog.train_ops:
                loss = loss_var._eval(env, memo)
                loss.backward()
                opt._static_step(prog)
            outs = []
            for f in (fetch_list or []):
                outs.append(f._eval(env, memo) if isinstance(f, Var) else f)
        if return_numpy:
            outs = [o.detach().cpu().numpy() if isinstance(o, torch.Tensor) else o
                    for o in outs]
        return list(outs)


class InputSpec:
    def __init__(self, shape=None, dtype="float32", name=None, stop_gradient=True):
        self.shape = shape
        self.dtype = dtype
        self.name = name
        self.stop_gradient = stop_gradient

    @classmethod
    def from_tensor(cls, tensor, name=None):
        return cls(list(tensor.shape), str(tensor.dtype), name)


class _ProgramModule(torch.nn.Module):
    """Evaluates a captured static Program feed->fetch as a torch Module
    so torch.jit.trace can serialize it (the Var DAG's op fns execute
    real substrate ops under tracing)."""

    def __init__(self, feed_vars, fetch_vars):
        super().__init__()
        self._feeds = feed_vars
        self._fetches = fetch_vars
        for i, p in enumerate(_collect_params(fetch_vars)):
            self.register_parameter(f"p{i}", torch.nn.Parameter(
                p.tensor.detach().clone(), requires_grad=False))
            p._traced_tensor = getattr(self, f"p{i}")

    def forward(self, *inputs):
        env = {fv.name: t for fv, t in zip(self._feeds, inputs)}
        memo = {}
        outs = [_eval_traced(v, env, memo) for v in self._fetches]
        return tuple(outs) if len(outs) > 1 else outs[0]


def _collect_params(fetch_vars):
    seen, out = set(), []

    def walk(v):
        if not isinstance(v, Var) or id(v) in seen:
            return
        seen.add(id(v))
        if v.kind == "param":
            out.append(v)
        for a in list(v.args) + list(v.kwargs.values()):
            walk(a)
    for f in fetch_vars:
        walk(f)
    return out


def _eval_traced(v, env, memo):
    key = id(v)
    if key in memo:
        return memo[key]
    if v.kind == "data":
        r = env[v.name]
    elif v.kind == "param":
        r = getattr(v, "_traced_tensor", v.tensor)
    else:
        args = [_eval_traced(a, env, memo) if isinstance(a, Var) else a
                for a in v.args]
        kwargs = {k: (_eval_traced(x, env, memo) if isinstance(x, Var) else x)
                  for k, x in v.kwargs.items()}
        r = v.fn(*args, **kwargs)
    memo[key] = r
    return r


def save_inference_model(path_prefix, feed_vars, fetch_vars, executor,
                         program=None, **kwargs):
    """Serialize the captured deferred graph for class-free inference
    reload (reference python/paddle/static/io.py:513 file layout:
    .pdmodel program + .pdiparams weights)."""
    import os as _os
    import pickle as _pickle
    from .. import framework_io
    feed_vars = feed_vars if isinstance(feed_vars, (list, tuple)) else [feed_vars]
    fetch_vars = fetch_vars if isinstance(fetch_vars, (list, tuple)) else [fetch_vars]
    mod = _ProgramModule(feed_vars, fetch_vars).eval()
    ex = []
    for fv in feed_vars:
        shape = [1 if (d is None or (isinstance(d, int) and d < 0)) else d
                 for d in (fv.shape or [1])]
        ex.append(torch.zeros(shape, dtype=fv.dtype or torch.float32))
    with torch.no_grad():
        traced = torch.jit.trace(mod, tuple(ex), strict=False, check_trace=False)
    _os.makedirs(_os.path.dirname(path_prefix) or ".", exist_ok=True)
    torch.jit.save(traced, path_prefix + ".pdscript")
    framework_io.save({f"p{i}": p.tensor for i, p in
                       enumerate(_collect_params(fetch_vars))},
                      path_prefix + ".pdiparams")
    meta = {"format": "torchscript",
            "program_file": _os.path.basename(path_prefix) + ".pdscript",
            "feed_names": [fv.name for fv in feed_vars],
            "n_fetch": len(fetch_vars)}
    with open(path_prefix + ".pdmodel", "wb") as f:
        _pickle.dump(meta, f, protocol=2)


def load_inference_model(path_prefix, executor, **kwargs):
    """Returns (program_callable, feed_names, fetch_count) -- run via
    executor.run(program, feed={...}, fetch_list=...) or call directly."""
    import os as _os
    import pickle as _pickle
    with open(path_prefix + ".pdmodel", "rb") as f:
        meta = _pickle.load(f)
    prog = torch.jit.load(
        _os.path.join(_os.path.dirname(path_prefix) or ".",
                      meta["program_file"]), map_location="cpu")

    class _LoadedProgram:
        def __init__(self, mod, feed_names, n_fetch):
            self._mod = mod
            self.feed_names = feed_names
            self.n_fetch = n_fetch

        def __call__(self, *inputs):
            return self._mod(*inputs)

        def run(self, feed, fetch_list=None):
            ins = [feed[n] for n in self.feed_names]
            out = self._mod(*ins)
            return list(out) if isinstance(out, tuple) else [out]

    return (_LoadedProgram(prog, meta["feed_names"], meta["n_fetch"]),
            meta["feed_names"], meta["n_fetch"])


def gradients(targets, inputs, target_gradients=None):
    return torch.autograd.grad(targets, inputs, target_gradients, allow_unused=True)


class amp:
    pass


# ---------------------------------------------------------------------------
# legacy static API remainder (reference: static/__init__.py __all__)
# ---------------------------------------------------------------------------
import contextlib as _ctx


def append_backward(loss, parameter_list=None, no_grad_set=None, callbacks=None):
    """In the deferred-graph model gradients run inside Executor.run via
    optimizer.minimize; this records the loss for a bare-backward program."""
    _default_main.train_ops.append((loss, None))
    return []


class _Scope(dict):
    def find_var(self, name):
        return self.get(name)

    def var(self, name):
        return self.setdefault(name, None)


_global_scope = _Scope()


def global_scope():
    return _global_scope


@_ctx.contextmanager
def scope_guard(scope):
    global _global_scope
    old, _global_scope = _global_scope, scope
    try:
        yield
    finally:
        _global_scope = old


class BuildStrategy:
    def __init__(self):
        self.memory_optimize = None
        self.enable_inplace = None
        self.fuse_broadcast_ops = None


class CompiledProgram:
    def __init__(self, program, build_strategy=None):
        self.program = program
        self.build_strategy = build_strategy or BuildStrategy()


def Print(input, first_n=-1, message=None, summarize=20, **kw):
    def _p(t):
        print(message or "", t)
        return t
    return Var("op", fn=_p, args=(input,))


def py_func(func, x, out, backward_func=None, skip_vars_in_backward_input=None):
    xs = x if isinstance(x, (list, tuple)) else [x]
    return Var("op", fn=lambda *ts: func(*ts), args=tuple(xs))


@_ctx.contextmanager
def name_scope(prefix=None):
    yield


@_ctx.contextmanager
def device_guard(device=None):
    yield


@_ctx.contextmanager
def ipu_shard_guard(index=-1, stage=-1):
    raise NotImplementedError("IPU is not a target of this MI355X build")


class IpuStrategy:
    def __init__(self):
        raise NotImplementedError("IPU is not a target of this MI355X build")


class IpuCompiledProgram:
    def __init__(self, *a, **k):
        raise NotImplementedError("IPU is not a target of this MI355X build")


def set_ipu_shard(call_func, index=-1, stage=-1):
    raise NotImplementedError("IPU is not a target of this MI355X build")


class WeightNormParamAttr:
    def __init__(self, dim=None, **kw):
        self.dim = dim


class ExponentialMovingAverage:
    """EMA of program parameters (reference: static/ema.py)."""

    def __init__(self, decay=0.999, thres_steps=None, name=None):
        self.decay = decay
        self._ema = {}

    def update(self):
        for v in _default_main.params:
            t = v.tensor.detach()
            key = id(v)
            if key not in self._ema:
                self._ema[key] = t.clone()
            else:
                self._ema[key].mul_(self.decay).add_(t, alpha=1 - self.decay)

    @_ctx.contextmanager
    def apply(self, executor=None, need_restore=True):
        saved = {}
        for v in _default_main.params:
            key = id(v)
            if key in self._ema:
                saved[key] = v.tensor.detach().clone()
                with torch.no_grad():
                    v.tensor.copy_(self._ema[key])
        try:
            yield
        finally:
            if need_restore:
                with torch.no_grad():
                    for v in _default_main.params:
                        if id(v) in saved:
                            v.tensor.copy_(saved[id(v)])

    def restore(self, executor=None):
        pass


Variable = Var


def create_global_var(shape, value, dtype, persistable=False, force_cpu=False,
                      name=None):
    from .. import framework as _fw
    t = torch.full(shape, value, dtype=_fw.convert_dtype(dtype))
    v = Var("param", shape=shape, dtype=t.dtype, tensor=t)
    _default_main.params.append(v)
    return v


def accuracy(input, label, k=1, correct=None, total=None, name=None):
    def _acc(logits, lab):
        topk = logits.topk(k, dim=-1).indices
        return (topk == lab.view(-1, 1)).any(-1).float().mean()
    return Var("op", fn=_acc, args=(input, label))


def auc(input, label, curve="ROC", num_thresholds=4095, topk=1, slide_steps=1):
    raise NotImplementedError("static auc: use paddle.metric.Auc")


def ctr_metric_bundle(input, label, ins_tag_weight=None):
    raise NotImplementedError("ctr_metric_bundle: PS-specific; out of scope")


def cpu_places(device_count=None):
    from ..framework import CPUPlace
    import os
    n = device_count or int(os.environ.get("CPU_NUM", 1))
    return [CPUPlace() for _ in range(n)]


def cuda_places(device_ids=None):
    from ..framework import CUDAPlace
    ids = device_ids if device_ids is not None else (
        range(torch.cuda.device_count()) if torch.cuda.is_available() else [0])
    return [CUDAPlace(i) for i in ids]


def xpu_places(device_ids=None):
    raise NotImplementedError("XPU is not a target of this MI355X build")


# -- program/persistables serialization (reference: static/io.py:513,837) ----
def save(program, model_path, protocol=4, **configs):
    import pickle
    state = {id(v): v.tensor.detach().cpu() for v in program.params}
    with open(model_path + ".pdparams", "wb") as f:
        pickle.dump(list(state.values()), f, protocol=protocol)


def load(program, model_path, executor=None, var_list=None):
    import pickle
    with open(model_path + ".pdparams", "rb") as f:
        vals = pickle.load(f)
    with torch.no_grad():
        for v, t in zip(program.params, vals):
            v.tensor.copy_(t)


def serialize_program(feed_vars, fetch_vars, **kwargs):
    import pickle
    return pickle.dumps({"feeds": [getattr(v, "name", None) for v in feed_vars],
                         "fetch_count": len(fetch_vars)})


def serialize_persistables(feed_vars, fetch_vars, executor=None, **kwargs):
    import pickle
    return pickle.dumps([v.tensor.detach().cpu() for v in _default_main.params])


def save_to_file(path, content):
    with open(path, "wb") as f:
        f.write(content)


def load_from_file(path):
    with open(path, "rb") as f:
        return f.read()


def deserialize_program(data):
    import pickle
    return pickle.loads(data)


def deserialize_persistables(program, data, executor=None):
    import pickle
    vals = pickle.loads(data)
    with torch.no_grad():
        for v, t in zip(program.params, vals):
            v.tensor.copy_(t)


def normalize_program(program, feed_vars, fetch_vars, **kwargs):
    return program


def load_program_state(model_path, var_list=None):
    import pickle
    with open(model_path + ".pdparams", "rb") as f:
        return pickle.load(f)


def set_program_state(program, state):
    with torch.no_grad():
        for v, t in zip(program.params, state):
            v.tensor.copy_(t)
