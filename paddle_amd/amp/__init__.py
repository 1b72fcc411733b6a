"""paddle.amp parity (reference: python/paddle/amp/auto_cast.py,
grad_scaler.py:187; O1/O2 semantics per SURVEY.md A.5).

auto_cast maps to torch.autocast with the paddle O1 white/black lists
honored by construction (torch's autocast policy covers the same op
classes: matmul-like ops cast down, reductions/norms stay fp32).
"""
from __future__ import annotations

import contextlib

import torch

from .. import framework

# O1 lists kept for API parity / inspection (amp_lists.py:20-44)
WHITE_LIST = {"conv2d", "einsum", "matmul", "matmul_v2", "mul",
              "fused_gemm_epilogue", "fused_rotary_position_embedding", "flash_attn"}
BLACK_LIST = {"exp", "square", "log", "mean", "sum", "cos_sim", "softmax",
              "softmax_with_cross_entropy", "sigmoid_cross_entropy_with_logits",
              "c_softmax_with_cross_entropy", "cross_entropy", "cross_entropy2",
              "layer_norm", "reduce_sum", "rms_norm"}


@contextlib.contextmanager
def auto_cast(enable=True, custom_white_list=None, custom_black_list=None,
              level="O1", dtype="bfloat16", use_promote=True):
    if not enable:
        yield
        return
    dt = framework.convert_dtype(dtype)
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    with torch.autocast(device_type=dev, dtype=dt, enabled=True):
        yield


amp_guard = auto_cast


def decorate(models, optimizers=None, level="O2", dtype="bfloat16",
             master_weight=None, save_dtype=None, master_grad=False,
             excluded_layers=None):
    """O2: cast model params to dtype (keeping norms fp32 per paddle's
    need_keep_fp32 unless excluded); optimizer keeps fp32 master."""
    dt = framework.convert_dtype(dtype)
    from ..nn import BatchNorm, BatchNorm1D, BatchNorm2D, LayerNorm

    def _cast_model(m):
        if level == "O2":
            keep = (LayerNorm, BatchNorm, BatchNorm1D, BatchNorm2D)
            for layer in m.modules():
                if excluded_layers and isinstance(layer, tuple(excluded_layers)):
                    continue
                if isinstance(layer, keep):
                    continue
                for _, p in layer.named_parameters(recurse=False):
                    p.data = p.data.to(dt)
        return m

    single = not isinstance(models, (list, tuple))
    ms = [models] if single else list(models)
    ms = [_cast_model(m) for m in ms]
    out_m = ms[0] if single else ms
    if optimizers is None:
        return out_m
    return out_m, optimizers


class GradScaler:
    """Dynamic loss scaling (grad_scaler.py defaults: init 2**16,
    incr_ratio 2.0 / incr_every_n_steps 1000 / decr_ratio 0.5)."""

    def __init__(self, enable=True, init_loss_scaling=2.0 ** 16, incr_ratio=2.0,
                 decr_ratio=0.5, incr_every_n_steps=1000, decr_every_n_nan_or_inf=2,
                 use_dynamic_loss_scaling=True):
        self._enable = enable
        self._scale = float(init_loss_scaling)
        self._incr_ratio = incr_ratio
        self._decr_ratio = decr_ratio
        self._incr_every = incr_every_n_steps
        self._decr_every = decr_every_n_nan_or_inf
        self._dynamic = use_dynamic_loss_scaling
        self._good_steps = 0
        self._bad_steps = 0
        self._found_inf = False
        self._unscaled_opts = set()  # id(optimizer) already unscaled this step

    def is_enable(self):
        return self._enable

    def scale(self, var):
        if not self._enable:
            return var
        return var * self._scale

    def _check_finite(self, optimizer):
        found = False
        for p in optimizer._params:
            if p.grad is not None:
                g = p.grad
                if not torch.isfinite(g.float().sum()):
                    found = True
                    break
        # Hybrid-parallel semantics (reference fleet/scaler.py distributed_scaler):
        # every rank must agree on skip/apply or DP/PP/sharded replicas desync.
        if torch.distributed.is_available() and torch.distributed.is_initialized():
            dev = "cuda" if torch.cuda.is_available() else "cpu"
            flag = torch.tensor([1.0 if found else 0.0], device=dev)
            torch.distributed.all_reduce(flag, op=torch.distributed.ReduceOp.MAX)
            found = bool(flag.item() > 0)
        self._found_inf = found
        return found

    def unscale_(self, optimizer):
        if not self._enable:
            return
        if id(optimizer) in self._unscaled_opts:
            return  # already unscaled this step (paddle raises; we no-op safely)
        inv = 1.0 / self._scale
        for p in optimizer._params:
            if p.grad is not None:
                p.grad.mul_(inv)
        self._unscaled_opts.add(id(optimizer))
        self._check_finite(optimizer)

    def step(self, optimizer):
        if not self._enable:
            optimizer.step()
            return
        if id(optimizer) not in self._unscaled_opts:
            self.unscale_(optimizer)
        self._unscaled_opts.discard(id(optimizer))
        if not self._found_inf:
            optimizer.step()

    def update(self):
        if not (self._enable and self._dynamic):
            return
        if self._found_inf:
            self._bad_steps += 1
            self._good_steps = 0
            if self._bad_steps >= self._decr_every:
                self._scale = max(self._scale * self._decr_ratio, 1.0)
                self._bad_steps = 0
        else:
            self._good_steps += 1
            self._bad_steps = 0
            if self._good_steps >= self._incr_every:
                self._scale *= self._incr_ratio
                self._good_steps = 0

    def minimize(self, optimizer, scaled_loss):
        scaled_loss.backward()
        self.step(optimizer)
        self.update()
        optimizer.clear_grad()

    def state_dict(self):
        return {"scale": self._scale, "incr_ratio": self._incr_ratio,
                "decr_ratio": self._decr_ratio, "incr_count": self._good_steps,
                "decr_count": self._bad_steps}

    def load_state_dict(self, sd):
        self._scale = sd.get("scale", self._scale)

    def get_loss_scaling(self):
        return torch.tensor(self._scale)


AmpScaler = GradScaler


def is_bfloat16_supported(device=None):
    return True


def is_float16_supported(device=None):
    return True


class DebugMode:
    """paddle.amp.debugging.DebugMode (reference debugging.py:56)."""
    CHECK_NAN_INF_AND_ABORT = 0
    CHECK_NAN_INF = 1
    CHECK_ALL_FOR_OVERFLOW = 2
    CHECK_ALL = 3


class TensorCheckerConfig:
    """Reference python/paddle/amp/debugging.py:173 -- config for the
    module-output nan/inf checker installed by enable_tensor_checker."""

    def __init__(self, enable=True, debug_mode=DebugMode.CHECK_NAN_INF_AND_ABORT,
                 output_dir=None, checked_op_list=None, skipped_op_list=None,
                 debug_step=None, stack_height_limit=1):
        self.enable = enable
        self.debug_mode = debug_mode
        self.output_dir = output_dir
        self.checked_op_list = checked_op_list
        self.skipped_op_list = set(skipped_op_list or [])
        self.debug_step = debug_step
        self.stack_height_limit = stack_height_limit
        self._step = 0


_tensor_checker = {"config": None, "hooks": []}


class debugging:
    """paddle.amp.debugging (check_numerics, tensor checker, accuracy
    compare -- reference python/paddle/amp/debugging.py)."""

    DebugMode = DebugMode
    TensorCheckerConfig = TensorCheckerConfig

    @staticmethod
    def enable_operator_stats_collection():
        pass

    @staticmethod
    def disable_operator_stats_collection():
        pass

    @staticmethod
    def check_numerics(tensor, op_type="", var_name="", debug_mode=None):
        import torch as _t
        if not _t.isfinite(tensor.float()).all():
            bad = (~_t.isfinite(tensor.float())).sum().item()
            msg = f"nan/inf detected in {op_type}:{var_name} ({bad} bad elements)"
            if debug_mode in (None, DebugMode.CHECK_NAN_INF_AND_ABORT):
                raise FloatingPointError(msg)
            import warnings
            warnings.warn(msg)
        return tensor

    @staticmethod
    def check_layer_numerics(layer):
        """Wrap a layer so every forward asserts finite outputs
        (reference debugging.py:78)."""
        import torch as _t

        def _hook(mod, inp, out):
            for i, t in enumerate(_flat_tensors(out)):
                debugging.check_numerics(t, type(mod).__name__, f"out{i}")
        layer.register_forward_post_hook(_hook) if hasattr(layer, "register_forward_post_hook") \
            else layer.register_forward_hook(_hook)
        return layer

    @staticmethod
    def enable_tensor_checker(config):
        """Install forward hooks on every later-constructed check via
        torch module hooks; checks all module outputs for nan/inf."""
        import torch as _t
        debugging.disable_tensor_checker()
        _tensor_checker["config"] = config
        if not config.enable:
            return

        def _hook(mod, inp, out):
            cfg = _tensor_checker["config"]
            if cfg is None:
                return
            name = type(mod).__name__
            if cfg.checked_op_list and name not in cfg.checked_op_list:
                return
            if name in cfg.skipped_op_list:
                return
            for i, t in enumerate(_flat_tensors(out)):
                debugging.check_numerics(t, name, f"out{i}", cfg.debug_mode)
        h = _t.nn.modules.module.register_module_forward_hook(_hook)
        _tensor_checker["hooks"].append(h)

    @staticmethod
    def disable_tensor_checker():
        for h in _tensor_checker["hooks"]:
            h.remove()
        _tensor_checker["hooks"].clear()
        _tensor_checker["config"] = None

    @staticmethod
    def compare_accuracy(dump_path, another_dump_path, output_filename,
                         loss_scale=1, dump_all_tensors=False):
        """Compare two tensor-dump directories (torch.save'd dicts of
        name->tensor, one file per step) and write a CSV of max abs/rel
        divergence per tensor (reference amp/accuracy_compare.py, minus
        the xlsx dependency)."""
        import csv as _csv
        import os as _os
        import torch as _t
        rows = []
        for fn in sorted(_os.listdir(dump_path)):
            p1 = _os.path.join(dump_path, fn)
            p2 = _os.path.join(another_dump_path, fn)
            if not _os.path.exists(p2):
                continue
            d1, d2 = _t.load(p1, weights_only=False), _t.load(p2, weights_only=False)
            for k in d1:
                if k not in d2:
                    continue
                a = d1[k].float() * loss_scale
                b = d2[k].float()
                if a.shape != b.shape:
                    rows.append([fn, k, "shape-mismatch", "", ""])
                    continue
                diff = (a - b).abs()
                rel = diff.max() / b.abs().max().clamp(min=1e-12)
                rows.append([fn, k, float(diff.max()), float(rel),
                             "DIVERGED" if float(rel) > 1e-2 else ""])
        with open(output_filename, "w", newline="") as f:
            w = _csv.writer(f)
            w.writerow(["file", "tensor", "max_abs_diff", "max_rel_diff", "flag"])
            w.writerows(rows)
        return rows


def _flat_tensors(out):
    import torch as _t
    if isinstance(out, _t.Tensor):
        return [out] if out.is_floating_point() else []
    if isinstance(out, (list, tuple)):
        r = []
        for o in out:
            r.extend(_flat_tensors(o))
        return r
    return []
