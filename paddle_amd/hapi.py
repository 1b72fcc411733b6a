"""paddle.Model high-level API (reference: python/paddle/hapi/model.py:1472
Model.fit/evaluate/predict + callbacks)."""
from __future__ import annotations

import time
from typing import List, Optional

import numpy as np
import torch

from . import io as pio
from . import metric as pmetric
from .framework_io import load as fload
from .framework_io import save as fsave


class Callback:
    def set_params(self, params):
        self.params = params

    def on_train_begin(self, logs=None):
        pass

    def on_train_end(self, logs=None):
        pass

    def on_epoch_begin(self, epoch, logs=None):
        pass

    def on_epoch_end(self, epoch, logs=None):
        pass

    def on_train_batch_begin(self, step, logs=None):
        pass

    def on_train_batch_end(self, step, logs=None):
        pass


class ProgBarLogger(Callback):
    def __init__(self, log_freq=10, verbose=1):
        self.log_freq = log_freq
        self.verbose = verbose

    def on_train_batch_end(self, step, logs=None):
        if self.verbose and step % self.log_freq == 0:
            items = ", ".join(f"{k}: {v:.4f}" if isinstance(v, float) else f"{k}: {v}"
                              for k, v in (logs or {}).items())
            print(f"step {step} - {items}")


class ModelCheckpoint(Callback):
    def __init__(self, save_freq=1, save_dir=None):
        self.save_freq = save_freq
        self.save_dir = save_dir

    def on_epoch_end(self, epoch, logs=None):
        if self.save_dir and epoch % self.save_freq == 0:
            self.model.save(f"{self.save_dir}/{epoch}")


class Model:
    def __init__(self, network, inputs=None, labels=None):
        self.network = network
        self._optimizer = None
        self._loss = None
        self._metrics: List = []

    def prepare(self, optimizer=None, loss=None, metrics=None, amp_configs=None):
        self._optimizer = optimizer
        self._loss = loss
        if metrics is not None:
            self._metrics = metrics if isinstance(metrics, (list, tuple)) else [metrics]
        return self

    def _run_one_batch(self, batch, train=True):
        if isinstance(batch, (list, tuple)) and len(batch) >= 2:
            x, y = batch[0], batch[1]
        else:
            x, y = batch, None
        out = self.network(x)
        logs = {}
        if self._loss is not None and y is not None:
            loss = self._loss(out, y)
            logs["loss"] = float(loss.detach().float().cpu())
            if train:
                loss.backward()
                self._optimizer.step()
                self._optimizer.clear_grad()
        for m in self._metrics:
            try:
                corr = m.compute(out, y)
                res = m.update(corr)
                logs[m.name()] = res
            except Exception:
                pass
        return logs

    def fit(self, train_data=None, eval_data=None, batch_size=1, epochs=1,
            eval_freq=1, log_freq=10, save_dir=None, save_freq=1, verbose=1,
            drop_last=False, shuffle=True, num_workers=0, callbacks=None,
            accumulate_grad_batches=1, num_iters=None):
        loader = train_data
        if not isinstance(train_data, pio.DataLoader) and train_data is not None and \
                not hasattr(train_data, "__iter__"):
            loader = pio.DataLoader(train_data, batch_size=batch_size, shuffle=shuffle,
                                    drop_last=drop_last, num_workers=num_workers)
        cbs = list(callbacks or [])
        if verbose:
            cbs.append(ProgBarLogger(log_freq, verbose))
        for cb in cbs:
            cb.model = self
            cb.on_train_begin()
        self.stop_training = False
        self.network.train()
        it = 0
        for epoch in range(epochs):
            for cb in cbs:
                cb.on_epoch_begin(epoch)
            for m in self._metrics:
                m.reset()
            for step, batch in enumerate(loader):
                logs = self._run_one_batch(batch, train=True)
                for cb in cbs:
                    cb.on_train_batch_end(step, logs)
                it += 1
                if num_iters is not None and it >= num_iters:
                    break
            for cb in cbs:
                cb.on_epoch_end(epoch, logs)
            if eval_data is not None and (epoch + 1) % eval_freq == 0:
                eval_logs = self.evaluate(eval_data, batch_size=batch_size,
                                          verbose=0)
                for cb in cbs:
                    cb.on_eval_end(eval_logs)
            if save_dir and (epoch + 1) % save_freq == 0:
                self.save(f"{save_dir}/{epoch}")
            if getattr(self, "stop_training", False):
                break                        # EarlyStopping & friends
        for cb in cbs:
            cb.on_train_end()

    @torch.no_grad()
    def evaluate(self, eval_data, batch_size=1, log_freq=10, verbose=1,
                 num_workers=0, callbacks=None, num_iters=None):
        loader = eval_data
        if not hasattr(eval_data, "__iter__"):
            loader = pio.DataLoader(eval_data, batch_size=batch_size)
        self.network.eval()
        for m in self._metrics:
            m.reset()
        total_loss, n = 0.0, 0
        for batch in loader:
            logs = self._run_one_batch(batch, train=False)
            if "loss" in logs:
                total_loss += logs["loss"]
                n += 1
        self.network.train()
        out = {"loss": total_loss / max(n, 1)}
        for m in self._metrics:
            out[m.name()] = m.accumulate()
        if verbose:
            print("Eval:", out)
        return out

    @torch.no_grad()
    def predict(self, test_data, batch_size=1, num_workers=0, stack_outputs=False,
                verbose=1, callbacks=None):
        loader = test_data
        if not hasattr(test_data, "__iter__"):
            loader = pio.DataLoader(test_data, batch_size=batch_size)
        self.network.eval()
        outs = []
        for batch in loader:
            x = batch[0] if isinstance(batch, (list, tuple)) else batch
            outs.append(self.network(x).cpu().numpy())
        self.network.train()
        if stack_outputs:
            return [np.concatenate(outs, 0)]
        return [outs]

    def save(self, path, training=True):
        fsave(self.network.state_dict(), path + ".pdparams")
        if training and self._optimizer is not None:
            fsave(self._optimizer.state_dict(), path + ".pdopt")

    def load(self, path, skip_mismatch=False, reset_optimizer=False):
        sd = fload(path + ".pdparams")
        self.network.set_state_dict(sd)
        import os
        if not reset_optimizer and self._optimizer is not None and \
                os.path.exists(path + ".pdopt"):
            self._optimizer.set_state_dict(fload(path + ".pdopt"))

    def parameters(self, *a, **kw):
        return self.network.parameters(*a, **kw)

    def summary(self, input_size=None, dtype=None):
        n_params = sum(p.numel() for p in self.network.parameters())
        print(f"Total params: {n_params:,}")
        return {"total_params": n_params}


class EarlyStopping(Callback):
    """Stop fit() when a monitored metric stops improving (reference:
    hapi/callbacks.py:EarlyStopping)."""

    def __init__(self, monitor="loss", mode="auto", patience=0, verbose=1,
                 min_delta=0, baseline=None, save_best_model=True):
        self.monitor = monitor
        self.patience = patience
        self.min_delta = abs(min_delta)
        self.baseline = baseline
        self.save_best_model = save_best_model
        if mode == "max" or (mode == "auto" and "acc" in monitor):
            self.better = lambda cur, best: cur > best + self.min_delta
            self.best = -float("inf")
        else:
            self.better = lambda cur, best: cur < best - self.min_delta
            self.best = float("inf")
        if baseline is not None:
            self.best = baseline
        self.wait = 0
        self.stopped_epoch = 0

    def on_eval_end(self, logs=None):
        logs = logs or {}
        cur = logs.get(self.monitor)
        if cur is None:
            return
        cur = float(cur[0] if isinstance(cur, (list, tuple)) else cur)
        if self.better(cur, self.best):
            self.best = cur
            self.wait = 0
        else:
            self.wait += 1
            if self.wait >= self.patience:
                self.model.stop_training = True


class LRScheduler(Callback):
    """Step the optimizer's LRScheduler per epoch/batch (reference:
    hapi/callbacks.py:LRScheduler)."""

    def __init__(self, by_step=False, by_epoch=True):
        self.by_step = by_step
        self.by_epoch = by_epoch

    def _sched(self):
        opt = getattr(self.model, "_optimizer", None)
        lr = getattr(opt, "_lr", None) if opt else None
        return lr if hasattr(lr, "step") else None

    def on_epoch_end(self, epoch, logs=None):
        s = self._sched()
        if self.by_epoch and s:
            s.step()

    def on_train_batch_end(self, step, logs=None):
        s = self._sched()
        if self.by_step and s:
            s.step()


class ReduceLROnPlateau(Callback):
    """Scale LR down when the monitored metric plateaus (reference:
    hapi/callbacks.py:ReduceLROnPlateau)."""

    def __init__(self, monitor="loss", factor=0.1, patience=10, verbose=1,
                 mode="auto", min_delta=1e-4, cooldown=0, min_lr=0):
        self.monitor = monitor
        self.factor = factor
        self.patience = patience
        self.cooldown = cooldown
        self.min_lr = min_lr
        self.min_delta = min_delta
        if mode == "max" or (mode == "auto" and "acc" in monitor):
            self.better = lambda cur, best: cur > best + min_delta
            self.best = -float("inf")
        else:
            self.better = lambda cur, best: cur < best - min_delta
            self.best = float("inf")
        self.wait = 0
        self.cooldown_counter = 0

    def on_eval_end(self, logs=None):
        logs = logs or {}
        cur = logs.get(self.monitor)
        if cur is None:
            return
        cur = float(cur[0] if isinstance(cur, (list, tuple)) else cur)
        if self.cooldown_counter > 0:
            self.cooldown_counter -= 1
            return
        if self.better(cur, self.best):
            self.best = cur
            self.wait = 0
            return
        self.wait += 1
        if self.wait >= self.patience:
            opt = getattr(self.model, "_optimizer", None)
            if opt is not None:
                new_lr = max(opt.get_lr() * self.factor, self.min_lr)
                opt.set_lr(new_lr)
            self.cooldown_counter = self.cooldown
            self.wait = 0


class VisualDL(Callback):
    """Scalar logging; VisualDL itself is not installed, so this writes a
    plain JSONL the visualdl UI can be pointed at later."""

    def __init__(self, log_dir="./log"):
        self.log_dir = log_dir
        self._f = None

    def _write(self, tag, step, logs):
        import json
        import os
        if self._f is None:
            os.makedirs(self.log_dir, exist_ok=True)
            self._f = open(f"{self.log_dir}/scalars.jsonl", "a")
        rec = {"tag": tag, "step": step}
        for k, v in (logs or {}).items():
            try:
                rec[k] = float(v[0] if isinstance(v, (list, tuple)) else v)
            except (TypeError, ValueError):
                pass
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()

    def on_train_batch_end(self, step, logs=None):
        self._write("train", step, logs)

    def on_eval_end(self, logs=None):
        self._write("eval", 0, logs)


class WandbCallback(Callback):
    def __init__(self, *a, **kw):
        raise RuntimeError("wandb is not installed in this image (no network "
                           "egress); use VisualDL for local JSONL logging")
