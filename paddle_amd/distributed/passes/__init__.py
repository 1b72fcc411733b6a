"""paddle.distributed.passes (reference: distributed/passes/__init__.py --
static-graph pass registry).  The PIR pass pipeline is replaced by eager
composition on this stack; the registry shape is kept so strategy code
that registers/queries passes keeps working (each pass is a no-op
callable unless a callable impl is registered)."""
from __future__ import annotations

_REGISTRY = {}


class PassContext:
    def __init__(self):
        self.attrs = {}

    def set_attr(self, k, v):
        self.attrs[k] = v

    def get_attr(self, k, default=None):
        return self.attrs.get(k, default)


class _Pass:
    def __init__(self, name, attrs=None, impl=None):
        self.name = name
        self.attrs = dict(attrs or {})
        self._impl = impl

    def apply(self, main_programs=None, startup_programs=None, context=None):
        if self._impl is not None:
            return self._impl(main_programs, startup_programs,
                              context or PassContext())
        return main_programs

    def set_attr(self, k, v):
        self.attrs[k] = v


def register_pass(name):
    def deco(impl):
        _REGISTRY[name] = impl
        return impl
    return deco


def new_pass(name, pass_attrs=None):
    return _Pass(name, pass_attrs, _REGISTRY.get(name))


class PassManager:
    def __init__(self, passes=None):
        self.passes = list(passes or [])

    def append(self, p):
        self.passes.append(p)

    def apply(self, main_programs=None, startup_programs=None):
        ctx = PassContext()
        out = main_programs
        for p in self.passes:
            out = p.apply(out, startup_programs, ctx)
        return out
