"""API-breadth tests: linalg, fft, sparse, auto_parallel (single + 2-rank)."""
import numpy as np
import pytest
import torch

import paddle_amd as paddle
from dist_util import run_dist


def test_linalg_ops():
    a = paddle.to_tensor(np.random.rand(4, 4).astype("float32") + np.eye(4, dtype="float32"))
    assert paddle.linalg.det(a).abs() > 0
    u, s, vh = paddle.linalg.svd(a)
    np.testing.assert_allclose((u @ torch.diag(s) @ vh).numpy(), a.numpy(), atol=1e-4)
    q, r = paddle.linalg.qr(a)
    np.testing.assert_allclose((q @ r).numpy(), a.numpy(), atol=1e-4)
    inv = paddle.linalg.inv(a)
    np.testing.assert_allclose((a @ inv).numpy(), np.eye(4), atol=1e-4)
    spd = a @ a.t() + 4 * torch.eye(4)
    c = paddle.linalg.cholesky(spd)
    np.testing.assert_allclose((c @ c.t()).numpy(), spd.numpy(), atol=1e-4)


def test_fft_roundtrip():
    x = paddle.randn([8, 16])
    y = paddle.fft.ifft(paddle.fft.fft(x))
    np.testing.assert_allclose(y.real.numpy(), x.numpy(), atol=1e-5)
    r = paddle.fft.irfft(paddle.fft.rfft(x), n=16)
    np.testing.assert_allclose(r.numpy(), x.numpy(), atol=1e-5)


def test_sparse_coo_csr():
    i = [[0, 1, 2], [2, 0, 1]]
    v = [1.0, 2.0, 3.0]
    coo = paddle.sparse.sparse_coo_tensor(i, v, shape=(3, 3))
    assert paddle.sparse.is_sparse_coo(coo)
    dense = paddle.sparse.to_dense(coo)
    assert dense[0, 2] == 1.0 and dense[1, 0] == 2.0
    csr = paddle.sparse.to_sparse_csr(dense)
    assert paddle.sparse.is_sparse_csr(csr)
    y = paddle.sparse.matmul(coo, torch.eye(3))
    np.testing.assert_allclose(y.to_dense().numpy() if y.layout != torch.strided
                               else y.numpy(), dense.numpy())


def test_process_mesh_single():
    mesh = paddle.distributed.ProcessMesh([[0, 1], [2, 3]], dim_names=["dp", "mp"])
    assert mesh.shape == [2, 2]
    assert mesh.process_ids == [0, 1, 2, 3]
    assert mesh.get_dim_size("mp") == 2


def test_shard_tensor_two_ranks():
    run_dist("""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed import ProcessMesh, Shard, Replicate, Partial
        from paddle_amd.distributed import shard_tensor, reshard
        r = paddle.distributed.get_rank()
        mesh = ProcessMesh([0, 1], dim_names=["x"])
        torch.manual_seed(0)
        full = torch.randn(8, 4)
        d = shard_tensor(full, mesh, [Shard(0)])
        assert d.shape == (4, 4)
        assert torch.allclose(d, full[r*4:(r+1)*4])
        back = reshard(d, mesh, [Replicate()])
        assert torch.allclose(back, full)
        # partial -> replicate == allreduce
        p = full.clone()
        p.placements = [Partial()]
        p.process_mesh = mesh
        red = reshard(p, mesh, [Replicate()])
        assert torch.allclose(red, 2 * full)
        print("rank", r, "dist tensor ok")
    """)


def test_top_level_export_parity_complete():
    """Every name in the reference's paddle.__all__ resolves here.
    (reference: python/paddle/__init__.py __all__, 418 names)"""
    import os
    import re
    ref = "/root/reference/python/paddle/__init__.py"
    if not os.path.exists(ref):
        import pytest
        pytest.skip("reference tree not mounted")
    src = open(ref).read()
    names = re.findall(r"'([^']+)'",
                       re.search(r"__all__ = \[(.*?)\]", src, re.S).group(1))
    import paddle_amd
    missing = [n for n in names if not hasattr(paddle_amd, n)]
    assert not missing, f"missing {len(missing)}: {missing[:20]}"


def test_submodule_export_parity_complete():
    """Every __all__ name of the reference's key submodules resolves here."""
    import importlib
    import os
    import re
    import pytest
    base = "/root/reference/python/paddle"
    if not os.path.isdir(base):
        pytest.skip("reference tree not mounted")
    for mod, ref in [("nn", "nn"), ("nn.functional", "nn/functional"),
                     ("optimizer", "optimizer"), ("distribution", "distribution"),
                     ("io", "io"), ("fft", "fft"), ("vision", "vision"),
                     ("autograd", "autograd"), ("signal", "signal"),
                     ("metric", "metric"), ("amp", "amp"),
                     ("utils", "utils"), ("static", "static"), ("jit", "jit"),
                     ("sparse", "sparse"), ("incubate", "incubate"),
                     ("text", "text"), ("device", "device"),
                     ("hub", "hapi/hub"),
                     ("vision.transforms", "vision/transforms"),
                     ("vision.models", "vision/models"),
                     ("vision.datasets", "vision/datasets"),
                     ("vision.ops", "vision/ops"),
                     ("audio.functional", "audio/functional"),
                     ("audio.features", "audio/features"),
                     ("audio.backends", "audio/backends"),
                     ("incubate.autograd", "incubate/autograd"),
                     ("incubate.optimizer", "incubate/optimizer"),
                     ("nn.utils", "nn/utils"),
                     ("nn.initializer", "nn/initializer"),
                     ("incubate.nn", "incubate/nn"),
                     ("callbacks", "callbacks"),
                     ("inference", "inference")]:
        try:
            src = open(f"{base}/{ref}/__init__.py").read()
        except FileNotFoundError:
            src = open(f"{base}/{ref}.py").read()
        m = re.search(r"__all__ = \[(.*?)\]", src, re.S)
        names = re.findall(r"'([^']+)'", m.group(1)) if m else []
        ours = importlib.import_module(f"paddle_amd.{mod}")
        missing = [n for n in names if not hasattr(ours, n)]
        assert not missing, f"{mod}: missing {missing}"


def test_long_tail_layers_forward():
    """Spot-check the wrapped long-tail layers actually run."""
    import torch
    import paddle_amd as paddle
    nn = paddle.nn
    x = torch.randn(2, 4, 8, 8)
    assert nn.CELU()(x).shape == x.shape
    assert nn.InstanceNorm2D(4)(x).shape == x.shape
    assert nn.AdaptiveMaxPool2D(2)(x).shape == (2, 4, 2, 2)
    assert nn.PixelShuffle(2)(x).shape == (2, 1, 16, 16)
    assert nn.ZeroPad2D(1)(x).shape == (2, 4, 10, 10)
    y = nn.Maxout(groups=2, axis=1)(x)
    assert y.shape == (2, 2, 8, 8)
    # RNN over a custom cell
    cell = nn.GRUCell(8, 16)
    rnn = nn.RNN(cell)
    seq = torch.randn(3, 5, 8)
    out, st = rnn(seq)
    assert out.shape == (3, 5, 16)
    birnn = nn.BiRNN(nn.GRUCell(8, 16), nn.GRUCell(8, 16))
    out2, _ = birnn(seq)
    assert out2.shape == (3, 5, 32)
    # functional spot checks
    F = paddle.nn.functional
    assert F.maxout(x, 2).shape == (2, 2, 8, 8)
    m = F.sequence_mask(torch.tensor([2, 4]), maxlen=5)
    assert m.shape == (2, 5) and int(m.sum()) == 6
    loss = F.sigmoid_focal_loss(torch.randn(4, 3), torch.randint(0, 2, (4, 3)).float())
    assert loss.dim() == 0


def test_tensor_method_parity_complete():
    """All 377 reference tensor_method_func names resolve as torch.Tensor
    attributes after patching (reference: python/paddle/tensor/__init__.py)."""
    import os
    import re
    import pytest
    import torch
    import paddle_amd  # noqa: F401 -- applies the method patch
    ref = "/root/reference/python/paddle/tensor/__init__.py"
    if not os.path.exists(ref):
        pytest.skip("reference tree not mounted")
    src = open(ref).read()
    names = sorted(set(re.findall(
        r"'([^']+)'",
        re.search(r"tensor_method_func = \[(.*?)\]", src, re.S).group(1))))
    t = torch.randn(2, 2)
    missing = []
    for n in names:
        try:
            ok = hasattr(t, n)
        except RuntimeError:
            ok = True  # exists, dtype-gated (e.g. .imag on real tensors)
        if not ok:
            missing.append(n)
    assert not missing, missing


def test_every_module_imports():
    """Every paddle_amd submodule imports cleanly (catches latent errors in
    rarely-exercised files)."""
    import importlib
    import pkgutil
    import paddle_amd
    bad = []
    for m in pkgutil.walk_packages(paddle_amd.__path__, "paddle_amd."):
        if m.name.endswith("__main__"):
            continue
        try:
            importlib.import_module(m.name)
        except Exception as e:  # pragma: no cover
            bad.append((m.name, repr(e)))
    assert not bad, bad


def test_vision_ops_behavior():
    import torch
    import paddle_amd.vision.ops as O
    boxes = torch.tensor([[0, 0, 10, 10], [1, 1, 11, 11], [50, 50, 60, 60.]])
    keep = O.nms(boxes, 0.5, torch.tensor([0.9, 0.8, 0.7]))
    assert keep.tolist() == [0, 2]
    x = torch.arange(64.).reshape(1, 1, 8, 8)
    assert O.roi_pool(x, torch.tensor([[0, 0, 4, 4.]]),
                      torch.tensor([1]), 2).shape == (1, 1, 2, 2)
    assert O.roi_align(x, torch.tensor([[0, 0, 4, 4.]]),
                       torch.tensor([1]), 2).shape == (1, 1, 2, 2)
    pb = torch.tensor([[0, 0, 10, 10.]])
    dec = O.box_coder(pb, [1., 1, 1, 1], torch.zeros(1, 4),
                      code_type="decode_center_size")
    assert torch.allclose(dec, pb, atol=1e-5)
    priors, var = O.prior_box(torch.zeros(1, 8, 4, 4), torch.zeros(1, 3, 32, 32),
                              min_sizes=[8.0])
    assert priors.shape[-1] == 4 and var.shape == priors.shape
