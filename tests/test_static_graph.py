"""paddle.static deferred-graph executor (reference: python/paddle/static/,
base/executor.py Executor.run feed/fetch semantics)."""
import numpy as np
import paddle_amd as paddle


def test_static_linear_regression_converges():
    paddle.enable_static()
    try:
        main = paddle.static.Program()
        startup = paddle.static.Program()
        with paddle.static.program_guard(main, startup):
            x = paddle.static.data("x", [None, 4])
            y = paddle.static.data("y", [None, 1])
            pred = paddle.static.nn.fc(x, 1)
            loss = paddle.mean((pred - y) * (pred - y))
            opt = paddle.optimizer.SGD(learning_rate=0.1)
            opt.minimize(loss)

            exe = paddle.static.Executor(paddle.CPUPlace())
            exe.run(paddle.static.default_startup_program())
            rng = np.random.default_rng(0)
            W = rng.normal(size=(4, 1)).astype("float32")
            first = last = None
            for _ in range(150):
                xb = rng.normal(size=(16, 4)).astype("float32")
                yb = (xb @ W + 0.1).astype("float32")
                (lv,) = exe.run(feed={"x": xb, "y": yb}, fetch_list=[loss])
                first = float(lv) if first is None else first
                last = float(lv)
            assert last < 0.05 * first
    finally:
        paddle.disable_state() if hasattr(paddle, "disable_state") else paddle.disable_static()


def test_static_fetch_without_train():
    paddle.enable_static()
    try:
        main = paddle.static.Program()
        with paddle.static.program_guard(main):
            a = paddle.static.data("a", [None, 3])
            b = paddle.static.data("b", [None, 3])
            c = a + b * 2.0
            exe = paddle.static.Executor(paddle.CPUPlace())
            (out,) = exe.run(feed={"a": np.ones((2, 3), "float32"),
                                   "b": np.ones((2, 3), "float32")},
                             fetch_list=[c])
            assert np.allclose(out, 3.0)
    finally:
        paddle.disable_static()


def test_static_mlp_classifier_mnistlike():
    paddle.enable_static()
    try:
        main = paddle.static.Program()
        with paddle.static.program_guard(main):
            import torch
            x = paddle.static.data("x", [None, 20])
            y = paddle.static.data("y", [None], "int64")
            h = paddle.static.nn.fc(x, 32, activation="relu")
            logits = paddle.static.nn.fc(h, 4)
            loss = torch.nn.functional.cross_entropy(logits, y)
            opt = paddle.optimizer.Adam(learning_rate=1e-2)
            opt.minimize(loss)
            exe = paddle.static.Executor(paddle.CPUPlace())
            exe.run(paddle.static.default_startup_program())
            rng = np.random.default_rng(1)
            centers = rng.normal(size=(4, 20)).astype("float32") * 3
            first = last = None
            for _ in range(120):
                lab = rng.integers(0, 4, size=64)
                xb = centers[lab] + rng.normal(size=(64, 20)).astype("float32")
                (lv,) = exe.run(feed={"x": xb.astype("float32"), "y": lab},
                                fetch_list=[loss])
                first = float(lv) if first is None else first
                last = float(lv)
            assert last < 0.5 * first
    finally:
        paddle.disable_static()


def test_jit_save_load_fresh_process_predict(tmp_path):
    """save -> fresh process -> load -> predict parity,
    no model class needed."""
    import subprocess
    import sys
    d = str(tmp_path)
    save_code = f'''
import torch, paddle_amd as paddle
from paddle_amd import nn
from paddle_amd.static import InputSpec
torch.manual_seed(0)
class M(nn.Layer):
    def __init__(self):
        super().__init__()
        self.fc1 = nn.Linear(8, 16)
        self.ln = nn.LayerNorm(16)
        self.fc2 = nn.Linear(16, 4)
    def forward(self, x):
        return self.fc2(torch.relu(self.ln(self.fc1(x))))
m = M().eval()
x = torch.randn(3, 8)
paddle.jit.save(m, r"{d}/model", input_spec=[InputSpec([None, 8])])
torch.save({{"x": x, "ref": m(x)}}, r"{d}/io.pt")
import pickle
assert pickle.load(open(r"{d}/model.pdmodel", "rb"))["format"] == "torchscript"
'''
    load_code = f'''
import torch, paddle_amd as paddle
tl = paddle.jit.load(r"{d}/model")
io = torch.load(r"{d}/io.pt", weights_only=False)
assert (tl(io["x"]) - io["ref"]).abs().max().item() < 1e-5
'''
    for code in (save_code, load_code):
        r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                           text=True)
        assert r.returncode == 0, r.stderr[-2000:]


def test_jit_save_gpt_traced(tmp_path):
    """The flagship model family itself traces (flash-attn, fused norms and
    CE all have substrate trace forms)."""
    import paddle_amd as paddle
    from paddle_amd.models.gpt import GPTForPretraining, PRESETS
    import torch
    m = GPTForPretraining(PRESETS["gpt3-tiny"]).eval()
    ids = torch.randint(0, 1024, (2, 16))
    ref = m(ids)
    paddle.jit.save(m, str(tmp_path / "gpt"), input_spec=[ids])
    tl = paddle.jit.load(str(tmp_path / "gpt"))
    out = tl(ids)
    torch.testing.assert_close(out, ref, atol=1e-4, rtol=1e-4)


def test_save_load_inference_model(tmp_path):
    import torch
    from paddle_amd import static
    prog = static.Program()
    start = static.Program()
    with static.program_guard(prog, start):
        x = static.data("x", [None, 6], "float32")
        y = static.nn.fc(x, 5, activation="relu")
        z = static.nn.fc(y, 3)
    exe = static.Executor()
    exe.run(start)
    feed_x = torch.randn(4, 6)
    ref = exe.run(prog, feed={"x": feed_x}, fetch_list=[z])[0]
    static.save_inference_model(str(tmp_path / "m"), [x], [z], exe)
    loaded, feeds, nf = static.load_inference_model(str(tmp_path / "m"), exe)
    out = loaded.run({"x": feed_x})[0]
    torch.testing.assert_close(torch.as_tensor(out), torch.as_tensor(ref),
                               atol=1e-5, rtol=1e-5)


def test_to_static_captures_and_replays():
    import torch
    import paddle_amd as paddle

    calls = {"n": 0}

    @paddle.jit.to_static
    def f(x, y):
        calls["n"] += 1
        return paddle.nn.functional.relu(x) @ y + 1

    x, y = torch.randn(4, 8), torch.randn(8, 2)
    ref = torch.relu(x) @ y + 1
    out1 = f(x, y)
    assert f.concrete_program is not None
    n_after_trace = calls["n"]
    out2 = f(x, y)                      # replayed, no python re-execution
    assert calls["n"] == n_after_trace
    assert torch.allclose(out1, ref) and torch.allclose(out2, ref)
    xg = x.clone().requires_grad_(True)
    f(xg, y).sum().backward()           # autograd through the capture
    assert xg.grad is not None
    f(torch.randn(3, 8), y)             # new signature -> retrace
    assert len(f._traces) == 2
