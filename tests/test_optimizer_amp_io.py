"""Optimizers vs torch references; GradScaler; .pdparams/.pdopt roundtrip;
LR schedulers (reference: optimizer.py / lr.py / grad_scaler.py / io.py)."""
import math
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
