"""Optimizers vs torch references; GradScaler; .pdparams/.pdopt roundtrip;
LR schedulers (reference: optimizer.py / lr.py / grad_scaler.py / io.py)."""
import math
import pickle
import os

import numpy as np
import pytest
import torch

import paddle_amd as paddle
from paddle_amd import nn


def _train_pair(opt_ours_fn, opt_ref_fn, steps=5):
    torch.manual_seed(0)
    m1 = nn.Linear(8, 8)
    m2 = nn.Linear(8, 8)
    m2.set_state_dict(m1.state_dict())
    o1 = opt_ours_fn(m1)
    o2 = opt_ref_fn(m2)
    x = paddle.randn([16, 8])
    for _ in range(steps):
        l1 = m1(x).square().mean()
        l1.backward()
        o1.step()
        o1.clear_grad()
        l2 = m2(x).square().mean()
        l2.backward()
        o2.step()
        o2.zero_grad()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        torch.testing.assert_close(p1, p2, rtol=1e-4, atol=1e-5)


def test_adamw_vs_torch():
    _train_pair(
        lambda m: paddle.optimizer.AdamW(learning_rate=1e-2, parameters=m.parameters(),
                                         weight_decay=0.01),
        lambda m: torch.optim.AdamW(m.parameters(), lr=1e-2, weight_decay=0.01))


def test_sgd_vs_torch():
    _train_pair(
        lambda m: paddle.optimizer.SGD(learning_rate=1e-2, parameters=m.parameters()),
        lambda m: torch.optim.SGD(m.parameters(), lr=1e-2))


def test_momentum_vs_torch():
    _train_pair(
        lambda m: paddle.optimizer.Momentum(learning_rate=1e-2, momentum=0.9,
                                            parameters=m.parameters()),
        lambda m: torch.optim.SGD(m.parameters(), lr=1e-2, momentum=0.9))


def test_lr_schedulers():
    s = paddle.optimizer.lr.CosineAnnealingDecay(0.1, T_max=10)
    vals = []
    for _ in range(11):
        vals.append(s())
        s.step()
    assert abs(vals[0] - 0.1) < 1e-9
    assert vals[-1] < 0.002
    w = paddle.optimizer.lr.LinearWarmup(0.1, warmup_steps=5, start_lr=0.0, end_lr=0.1)
    assert w() == 0.0
    for _ in range(5):
        w.step()
    assert abs(w() - 0.1) < 1e-9
    n = paddle.optimizer.lr.NoamDecay(d_model=512, warmup_steps=100)
    n.step()
    assert n() > 0


def test_scheduler_drives_optimizer():
    m = nn.Linear(2, 2)
    sched = paddle.optimizer.lr.StepDecay(0.1, step_size=1, gamma=0.5)
    opt = paddle.optimizer.SGD(learning_rate=sched, parameters=m.parameters())
    assert opt.get_lr() == 0.1
    sched.step()
    assert opt.get_lr() == 0.05


def test_grad_scaler_flow():
    m = nn.Linear(4, 4)
    opt = paddle.optimizer.SGD(learning_rate=0.1, parameters=m.parameters())
    scaler = paddle.amp.GradScaler(init_loss_scaling=2.0)
    loss = m(paddle.randn([4, 4])).mean()
    scaled = scaler.scale(loss)
    assert abs(float(scaled) - 2 * float(loss)) < 1e-6
    scaled.backward()
    scaler.step(opt)
    scaler.update()
    opt.clear_grad()


def test_grad_scaler_inf_skips_step():
    m = nn.Linear(2, 2)
    w0 = m.weight.detach().clone()
    opt = paddle.optimizer.SGD(learning_rate=1.0, parameters=m.parameters())
    scaler = paddle.amp.GradScaler(init_loss_scaling=4.0, decr_every_n_nan_or_inf=1)
    loss = m(paddle.randn([2, 2])).mean()
    scaler.scale(loss).backward()
    m.weight.grad[0, 0] = float("inf")
    scaler.step(opt)
    scaler.update()
    torch.testing.assert_close(m.weight.detach(), w0)
    assert scaler._scale == 2.0  # halved after inf


def test_save_load_pdparams(tmp_path):
    m = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 2))
    p = str(tmp_path / "model.pdparams")
    paddle.save(m.state_dict(), p)
    loaded = paddle.load(p)
    m2 = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 2))
    m2.set_state_dict(loaded)
    x = paddle.randn([2, 4])
    torch.testing.assert_close(m(x), m2(x))


def test_save_load_bf16(tmp_path):
    t = {"w": torch.randn(4, 4).bfloat16(), "step": 7}
    p = str(tmp_path / "x.pdparams")
    paddle.save(t, p)
    back = paddle.load(p)
    assert back["w"].dtype == torch.bfloat16
    torch.testing.assert_close(back["w"].float(), t["w"].float())
    assert back["step"] == 7


def test_optimizer_state_dict_roundtrip(tmp_path):
    m = nn.Linear(4, 4)
    opt = paddle.optimizer.AdamW(learning_rate=1e-3, parameters=m.parameters())
    loss = m(paddle.randn([2, 4])).mean()
    loss.backward()
    opt.step()
    p = str(tmp_path / "o.pdopt")
    paddle.save(opt.state_dict(), p)
    opt2 = paddle.optimizer.AdamW(learning_rate=1e-3, parameters=m.parameters())
    opt2.set_state_dict(paddle.load(p))
    assert opt2._step_count == 1


def test_amp_autocast_cpu_noop():
    with paddle.amp.auto_cast(dtype="bfloat16"):
        x = paddle.randn([2, 2])
        y = paddle.matmul(x, x)
    assert y is not None


def test_async_save(tmp_path):
    p = str(tmp_path / "a.pdparams")
    t = paddle.async_save({"x": torch.ones(3)}, p)
    paddle.framework_io.clear_async_save_task_queue()
    assert os.path.exists(p)
    assert paddle.load(p)["x"].sum() == 3


def test_fused_linear_param_grad_add_accumulates():
    import torch
    from paddle_amd.ops import functional as hot
    torch.manual_seed(0)
    x = torch.randn(6, 4)
    dy = torch.randn(6, 3)
    dw = torch.ones(4, 3)
    db = torch.ones(3)
    dw2, db2 = hot.fused_linear_param_grad_add(x, dy, dw, db)
    assert dw2 is dw and db2 is db
    torch.testing.assert_close(dw, 1.0 + x.t() @ dy)
    torch.testing.assert_close(db, 1.0 + dy.sum(0))


def test_long_tail_optimizers_state_roundtrip():
    """Each torch-engine optimizer trains and round-trips its state dict."""
    import torch
    import paddle_amd as paddle
    for cls in ("Adagrad", "Adamax", "RAdam", "NAdam", "RMSProp", "Adadelta",
                "Rprop", "ASGD"):
        p = torch.nn.Parameter(torch.randn(6))
        opt = getattr(paddle.optimizer, cls)(learning_rate=0.01,
                                             parameters=[p])
        for _ in range(2):
            (p * p).sum().backward()
            opt.step()
            opt.clear_grad()
        sd = opt.state_dict()
        p2 = torch.nn.Parameter(p.detach().clone())
        opt2 = getattr(paddle.optimizer, cls)(learning_rate=0.01,
                                              parameters=[p2])
        (p2 * p2).sum().backward()
        opt2.step()          # instantiate state
        opt2.clear_grad()
        opt2.set_state_dict(sd)
        with torch.no_grad():
            p2.copy_(p)   # state dict restores optimizer state, not params
        for o, q in ((opt, p), (opt2, p2)):
            (q * q).sum().backward()
            o.step()
            o.clear_grad()
        torch.testing.assert_close(p.detach(), p2.detach(), atol=1e-6,
                                   rtol=1e-6)


def test_static_ema_apply_restore():
    import torch
    import paddle_amd as paddle
    paddle.enable_static()
    try:
        main = paddle.static.Program()
        with paddle.static.program_guard(main):
            w = paddle.static.create_parameter([4])
            for init in main.initializers:
                init()
            main.initializers = []
            ema = paddle.static.ExponentialMovingAverage(decay=0.5)
            with torch.no_grad():
                w.tensor.fill_(1.0)
            ema.update()
            with torch.no_grad():
                w.tensor.fill_(3.0)
            ema.update()                       # ema = 0.5*1 + 0.5*3 = 2
            with ema.apply():
                assert torch.allclose(w.tensor, torch.full((4,), 2.0))
            assert torch.allclose(w.tensor, torch.full((4,), 3.0))  # restored
    finally:
        paddle.disable_static()


def test_safetensors_roundtrip(tmp_path):
    import torch
    import paddle_amd as paddle
    m = paddle.nn.Linear(4, 3)
    path = str(tmp_path / "w.safetensors")
    paddle.save_safetensors(m.state_dict(), path)
    back = paddle.load_safetensors(path)
    for k, v in m.state_dict().items():
        torch.testing.assert_close(back[k], v.detach())
    import pytest
    with pytest.raises(ValueError):
        paddle.save_safetensors({"step": 3}, str(tmp_path / "bad.safetensors"))


def test_gradscaler_unscale_then_step_single_unscale():
    # unscale_ followed by step() must not divide by scale twice.
    m = nn.Linear(4, 4)
    opt = paddle.optimizer.SGD(learning_rate=1.0, parameters=m.parameters())
    scaler = paddle.amp.GradScaler(init_loss_scaling=2.0 ** 8)
    x = paddle.randn([2, 4])
    loss = m(x).sum()
    scaled = scaler.scale(loss)
    scaled.backward()
    scaler.unscale_(opt)  # user-side unscale for grad clipping
    g_after_unscale = [p.grad.clone() for p in m.parameters()]
    scaler.step(opt)      # must NOT unscale again
    # recompute reference grads at scale 1
    m2 = nn.Linear(4, 4)
    with torch.no_grad():
        for p2, p in zip(m2.parameters(), m.parameters()):
            pass  # shapes match; we only compare grad scale consistency
    # grads stored on m were unscaled exactly once: loss grad of sum() is all-ones-ish
    # check: sum-of-abs of grads is within 2x band of the re-derived unscaled grads
    m.zero_grad()
    loss2 = m(x).sum()
    loss2.backward()
    for g_once, p in zip(g_after_unscale, m.parameters()):
        torch.testing.assert_close(g_once, p.grad, atol=1e-5, rtol=1e-5)


def test_gradscaler_double_unscale_is_noop():
    m = nn.Linear(3, 3)
    opt = paddle.optimizer.SGD(learning_rate=0.1, parameters=m.parameters())
    scaler = paddle.amp.GradScaler(init_loss_scaling=16.0)
    loss = m(paddle.randn([2, 3])).sum()
    scaler.scale(loss).backward()
    scaler.unscale_(opt)
    g1 = [p.grad.clone() for p in m.parameters()]
    scaler.unscale_(opt)  # second call: no-op, not another /16
    for a, p in zip(g1, m.parameters()):
        torch.testing.assert_close(a, p.grad)


def test_bf16_pdparams_reference_interop(tmp_path):
    """Byte-compat: bf16 tensors must pickle as the reference's own form --
    a plain (name, uint16 ndarray) tuple with NO repo-private classes
    (reference python/paddle/framework/io.py:425 reduce_varbase)."""
    import pickletools, io as _io
    w = torch.randn(4, 5).bfloat16()
    p = str(tmp_path / "m.pdparams")
    paddle.save({"w": w, "b": torch.ones(3)}, p)
    raw = open(p, "rb").read()
    # scan pickle opcodes: no GLOBAL/STACK_GLOBAL may reference paddle_amd
    for op, arg, _ in pickletools.genops(raw):
        if op.name in ("GLOBAL", "STACK_GLOBAL") and arg and "paddle_amd" in str(arg):
            raise AssertionError(f"repo-private class in pickle: {arg}")
    # unpickle with plain pickle (what the reference's loader does first)
    obj = pickle.loads(raw)
    assert isinstance(obj["w"], tuple) and len(obj["w"]) == 2
    name, arr = obj["w"]
    assert isinstance(name, str) and isinstance(arr, np.ndarray)
    assert arr.dtype == np.uint16  # uint16-means-bf16 convention
    # and our loader restores bf16 values exactly
    back = paddle.load(p)
    assert back["w"].dtype == torch.bfloat16
    torch.testing.assert_close(back["w"].float(), w.float())
    assert back["b"].dtype == torch.float32


def test_load_reference_produced_bf16_fixture(tmp_path):
    """Simulate a file written by the reference: state values are
    (name, ndarray) tuples, bf16 as raw uint16 bits."""
    w = torch.randn(6).bfloat16()
    fixture = {"w": ("linear_0.w_0", w.view(torch.uint16).numpy()),
               "s": ("scalar_0", np.float64(3.5))}
    p = str(tmp_path / "ref.pdparams")
    with open(p, "wb") as f:
        pickle.dump(fixture, f, protocol=2)
    back = paddle.load(p)
    assert back["w"].dtype == torch.bfloat16
    torch.testing.assert_close(back["w"].float(), w.float())


def test_flash_attention_returns_lse_cpu():
    # CPU fallback path: return_softmax_lse must hand back the real LSE
    from paddle_amd.ops import functional as F
    q = torch.randn(2, 8, 4, 16)  # [b, s, h, d] bshd layout
    k = torch.randn(2, 8, 4, 16)
    v = torch.randn(2, 8, 4, 16)
    o, lse = F.flash_attention(q, k, v, causal=True, return_softmax_lse=True)
    assert lse is not None and lse.shape == (2, 4, 8)  # [b, h, sq]
    # verify against direct logsumexp of scaled causal scores
    s = torch.einsum("bshd,bthd->bhst", q, k) / math.sqrt(16)
    mask = torch.ones(8, 8, dtype=torch.bool).tril()
    s = s.masked_fill(~mask, float("-inf"))
    torch.testing.assert_close(lse, torch.logsumexp(s, -1), atol=1e-4, rtol=1e-4)
