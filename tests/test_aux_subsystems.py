"""Aux subsystems: profiler, hapi Model, distribution, launch CLI,
flags, metrics, distributed checkpoint (reference SURVEY.md §5)."""
import os
import subprocess
import sys

import numpy as np
import pytest
import torch

import paddle_amd as paddle
from paddle_amd import nn

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_profiler_records(tmp_path):
    with paddle.profiler.Profiler() as prof:
        x = paddle.randn([64, 64])
        for _ in range(3):
            x = paddle.matmul(x, x)
            prof.step()
    prof.export(str(tmp_path / "trace.json"))
    assert (tmp_path / "trace.json").exists()


def test_record_event():
    with paddle.profiler.RecordEvent("my_op"):
        paddle.ones([2, 2]).sum()


def test_hapi_model_fit_and_save(tmp_path):
    from paddle_amd.io import TensorDataset
    paddle.seed(0)
    x = torch.randn(64, 4)
    y = (x.sum(-1) > 0).long()
    ds = TensorDataset([x, y])
    net = nn.Sequential(nn.Linear(4, 16), nn.ReLU(), nn.Linear(16, 2))
    model = paddle.Model(net)
    opt = paddle.optimizer.Adam(learning_rate=1e-2, parameters=net.parameters())
    model.prepare(opt, nn.CrossEntropyLoss(), paddle.metric.Accuracy())
    model.fit(ds, batch_size=16, epochs=2, verbose=0)
    res = model.evaluate(ds, batch_size=16, verbose=0)
    assert res["acc"] > 0.6
    model.save(str(tmp_path / "m"))
    assert (tmp_path / "m.pdparams").exists()
    model2 = paddle.Model(nn.Sequential(nn.Linear(4, 16), nn.ReLU(), nn.Linear(16, 2)))
    model2.prepare(None, nn.CrossEntropyLoss())
    model2.load(str(tmp_path / "m"), reset_optimizer=True)


def test_distribution_api():
    n = paddle.distribution.Normal(0.0, 1.0)
    s = n.sample([100])
    assert s.shape == (100,)
    lp = n.log_prob(paddle.to_tensor(0.0))
    np.testing.assert_allclose(float(lp), -0.9189385, rtol=1e-5)
    u = paddle.distribution.Uniform(0.0, 2.0)
    assert 0 <= float(u.sample([1])) <= 2
    c = paddle.distribution.Categorical(logits=[0.0, 0.0])
    assert int(c.sample([1])) in (0, 1)
    kl = paddle.distribution.kl_divergence(n, paddle.distribution.Normal(1.0, 1.0))
    np.testing.assert_allclose(float(kl), 0.5, rtol=1e-5)


def test_launch_cli_single_proc(tmp_path):
    script = tmp_path / "train.py"
    script.write_text(
        "import os, sys\n"
        "sys.path.insert(0, %r)\n"
        "import paddle_amd as paddle\n"
        "assert os.environ['WORLD_SIZE'] == '2'\n"
        "paddle.distributed.init_parallel_env()\n"
        "import torch\n"
        "t = torch.ones(2)\n"
        "paddle.distributed.all_reduce(t)\n"
        "assert t.tolist() == [2.0, 2.0]\n"
        "print('rank', os.environ['RANK'], 'ok')\n" % REPO)
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""
    env["PYTHONPATH"] = REPO
    from dist_util import free_port
    r = subprocess.run([sys.executable, "-m", "paddle_amd.distributed.launch",
                        "--nproc_per_node", "2", "--master_port", str(free_port()),
                        "--log_dir", str(tmp_path / "logs"), str(script)],
                       env=env, capture_output=True, text=True, timeout=180,
                       cwd=str(tmp_path))
    assert r.returncode == 0, r.stdout + r.stderr
    assert (tmp_path / "logs" / "workerlog.1").exists()


def test_flags_api():
    paddle.set_flags({"FLAGS_check_nan_inf": True})
    assert paddle.get_flags("FLAGS_check_nan_inf")["FLAGS_check_nan_inf"] is True
    paddle.set_flags({"FLAGS_check_nan_inf": False})


def test_amp_debugging_check_numerics():
    t = paddle.to_tensor([1.0, float("nan")])
    with pytest.raises(FloatingPointError):
        paddle.amp.debugging.check_numerics(t, "op", "var")


def test_dist_checkpoint_single(tmp_path):
    from paddle_amd.distributed.checkpoint import load_state_dict, save_state_dict
    sd = {"w": torch.randn(4, 4), "step": 3}
    save_state_dict(sd, str(tmp_path / "ckpt"))
    target = {"w": torch.zeros(4, 4), "step": 0}
    out = load_state_dict(target, str(tmp_path / "ckpt"))
    torch.testing.assert_close(out["w"], sd["w"])
    assert out["step"] == 3


def test_device_stream_event_api():
    s = paddle.device.Stream() if torch.cuda.is_available() else None
    e = paddle.device.Event() if torch.cuda.is_available() else None
    assert paddle.device.get_device() in ("cpu",) or "gpu" in paddle.device.get_device()


def test_vision_lenet_and_transforms():
    from paddle_amd.vision.models import LeNet
    from paddle_amd.vision.transforms import Compose, Normalize, ToTensor
    t = Compose([ToTensor(), Normalize([0.5], [0.5])])
    img = np.random.randint(0, 255, (28, 28), dtype=np.uint8)
    x = t(img)
    assert x.shape == (1, 28, 28)
    m = LeNet()
    out = m(x.unsqueeze(0))
    assert out.shape == (1, 10)


# -- incubate.asp / autotune -------------------------------------------------
def test_asp_prune_and_decorate():
    import torch
    import paddle_amd as paddle
    from paddle_amd.incubate import asp
    m = paddle.nn.Sequential(paddle.nn.Linear(16, 32), paddle.nn.Linear(32, 8))
    pruned = asp.prune_model(m)
    assert len(pruned) == 2
    opt = asp.decorate(paddle.optimizer.AdamW(learning_rate=1e-3,
                                              parameters=m.parameters()))
    loss = m(torch.randn(4, 16)).pow(2).mean()
    loss.backward()
    opt.step()
    opt.clear_grad()
    for p in m.parameters():
        if p.dim() == 2:
            assert asp.check_sparsity(p)          # still 2:4 after a step
            assert 0.3 < asp.calculate_density(p) <= 0.5


def test_autotune_config_roundtrip():
    from paddle_amd.incubate import autotune
    cfg = autotune.set_config({"kernel": {"enable": False},
                               "dataloader": {"tuning_steps": 7}})
    assert cfg["dataloader"]["tuning_steps"] == 7
    assert autotune.get_config()["kernel"]["enable"] is False


def test_extension_imports_when_built():
    """If the in-tree _C.so exists it must import cleanly -- catches
    undefined-symbol link errors on the CPU box before any GPU run."""
    import os
    import paddle_amd
    so = os.path.join(os.path.dirname(paddle_amd.__file__), "_C.so")
    if os.path.exists(so):
        import paddle_amd._C  # noqa: F401


def test_early_stopping_and_reduce_lr():
    import paddle_amd as paddle
    from paddle_amd.callbacks import EarlyStopping, ReduceLROnPlateau

    class _FakeModel:
        stop_training = False

    es = EarlyStopping(monitor="loss", patience=2, mode="min")
    es.model = _FakeModel()
    for loss in (1.0, 0.5, 0.6, 0.7, 0.8):
        es.on_eval_end({"loss": loss})
    assert es.model.stop_training

    m = paddle.nn.Linear(2, 2)
    opt = paddle.optimizer.SGD(learning_rate=1.0, parameters=m.parameters())

    class _M2:
        _optimizer = opt
    rl = ReduceLROnPlateau(monitor="loss", factor=0.5, patience=1, min_delta=0)
    rl.model = _M2()
    for loss in (1.0, 1.0, 1.0):
        rl.on_eval_end({"loss": loss})
    # plateau at evals 2 and 3 -> two 0.5x reductions
    assert abs(opt.get_lr() - 0.25) < 1e-9


def test_visualdl_writes_jsonl(tmp_path):
    import json
    from paddle_amd.callbacks import VisualDL
    v = VisualDL(log_dir=str(tmp_path))
    v.on_train_batch_end(1, {"loss": 0.5})
    v.on_eval_end({"acc": [0.9]})
    lines = [json.loads(l) for l in open(tmp_path / "scalars.jsonl")]
    assert lines[0]["loss"] == 0.5 and lines[1]["acc"] == 0.9


def test_model_fit_with_early_stopping():
    import torch
    import paddle_amd as paddle
    from paddle_amd.callbacks import EarlyStopping

    class DS(paddle.io.Dataset):
        def __getitem__(self, i):
            torch.manual_seed(i)
            x = torch.randn(4)
            return x, x.sum().reshape(1)

        def __len__(self):
            return 16

    net = paddle.nn.Linear(4, 1)
    model = paddle.Model(net)
    opt = paddle.optimizer.SGD(learning_rate=0.0, parameters=net.parameters())
    model.prepare(opt, paddle.nn.MSELoss())
    es = EarlyStopping(monitor="loss", patience=1, mode="min")
    model.fit(DS(), eval_data=DS(), epochs=10, batch_size=4, verbose=0,
              callbacks=[es])
    # lr=0 -> eval loss constant -> early stop long before 10 epochs
    assert model.stop_training


def test_comm_desync_checker():
    """Desync checker catches ranks issuing different collectives."""
    import sys
    sys.path.insert(0, "tests")
    from dist_util import run_dist
    run_dist("""
        import torch
        import paddle_amd as paddle
        from paddle_amd.distributed import comm_check
        paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        comm_check.enable()
        t = torch.ones(4)
        paddle.distributed.all_reduce(t)
        assert comm_check.verify()           # in sync
        # now diverge: rank 0 broadcasts a different shape
        if r == 0:
            paddle.distributed.broadcast(torch.ones(8), src=0)
        else:
            paddle.distributed.broadcast(torch.ones(4, 2), src=0)
        try:
            comm_check.verify()
            raise SystemExit("desync not detected")
        except RuntimeError as e:
            assert "desync" in str(e)
        comm_check.disable()
        print("rank", r, "desync checker ok")
    """, world_size=2)


def test_tensor_checker_and_compare_accuracy(tmp_path):
    import torch
    import paddle_amd as paddle
    dbg = paddle.amp.debugging
    cfg = dbg.TensorCheckerConfig(enable=True)
    dbg.enable_tensor_checker(cfg)
    m = torch.nn.Linear(4, 4)
    m(torch.randn(2, 4))
    m.weight.data[0, 0] = float("nan")
    import pytest as _pt
    with _pt.raises(FloatingPointError):
        m(torch.randn(2, 4))
    dbg.disable_tensor_checker()
    m(torch.randn(2, 4))  # no raise once disabled

    d1 = tmp_path / "a"; d2 = tmp_path / "b"
    d1.mkdir(); d2.mkdir()
    torch.save({"w": torch.ones(3)}, d1 / "s0.pt")
    torch.save({"w": torch.ones(3) * 1.5}, d2 / "s0.pt")
    rows = dbg.compare_accuracy(str(d1), str(d2), str(tmp_path / "cmp.csv"))
    assert rows and rows[0][4] == "DIVERGED"
