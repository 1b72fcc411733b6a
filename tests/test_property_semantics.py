"""Property-based semantics tests (hypothesis): paddle tensor-op behavior
vs numpy oracles over randomized shapes/values.  SURVEY §4 lists
randomized-input testing as part of the reference's op test strategy."""
import numpy as np
import pytest
import torch

import paddle_amd as paddle

try:
    from hypothesis import given, settings, strategies as st
    HAVE_HYP = True
except Exception:  # pragma: no cover
    HAVE_HYP = False

pytestmark = pytest.mark.skipif(not HAVE_HYP, reason="hypothesis absent")

shapes = st.lists(st.integers(1, 5), min_size=1, max_size=4)


@settings(max_examples=40, deadline=None)
@given(shape=shapes, data=st.data())
def test_reshape_roundtrip(shape, data):
    x = paddle.randn(shape)
    flat = paddle.reshape(x, [-1])
    back = paddle.reshape(flat, shape)
    assert torch.equal(back, x)
    # 0-rule: copy each input dim
    zero_shape = [0] * len(shape)
    assert tuple(paddle.reshape(x, zero_shape).shape) == tuple(shape)


@settings(max_examples=40, deadline=None)
@given(shape=shapes, data=st.data())
def test_reductions_match_numpy(shape, data):
    axis = data.draw(st.integers(0, len(shape) - 1))
    keep = data.draw(st.booleans())
    x = paddle.randn(shape)
    n = x.numpy()
    np.testing.assert_allclose(paddle.sum(x, axis=axis, keepdim=keep).numpy(),
                               n.sum(axis=axis, keepdims=keep), rtol=1e-5,
                               atol=1e-5)
    np.testing.assert_allclose(paddle.max(x, axis=axis, keepdim=keep).numpy(),
                               n.max(axis=axis, keepdims=keep), rtol=1e-6)
    np.testing.assert_allclose(paddle.mean(x, axis=axis, keepdim=keep).numpy(),
                               n.mean(axis=axis, keepdims=keep), rtol=1e-5,
                               atol=1e-5)


@settings(max_examples=30, deadline=None)
@given(shape=st.lists(st.integers(1, 6), min_size=2, max_size=3), data=st.data())
def test_concat_split_inverse(shape, data):
    axis = data.draw(st.integers(0, len(shape) - 1))
    k = data.draw(st.integers(1, 3))
    xs = [paddle.randn(shape) for _ in range(k)]
    cat = paddle.concat(xs, axis=axis)
    parts = paddle.split(cat, k, axis=axis)
    for a, b in zip(parts, xs):
        assert torch.equal(a, b)


@settings(max_examples=30, deadline=None)
@given(shape=shapes)
def test_save_load_roundtrip(tmp_path_factory, shape):
    import os
    d = tmp_path_factory.mktemp("ckpt")
    x = paddle.randn(shape)
    sd = {"w": x, "b": x.to(torch.bfloat16), "n": 3}
    p = os.path.join(str(d), "m.pdparams")
    paddle.save(sd, p)
    out = paddle.load(p)
    assert torch.equal(out["w"], x)
    assert out["b"].dtype == torch.bfloat16
    assert torch.equal(out["b"], x.to(torch.bfloat16))
    assert out["n"] == 3


@settings(max_examples=30, deadline=None)
@given(data=st.data())
def test_transpose_perm(data):
    nd = data.draw(st.integers(2, 4))
    shape = data.draw(st.lists(st.integers(1, 4), min_size=nd, max_size=nd))
    perm = data.draw(st.permutations(list(range(nd))))
    x = paddle.randn(shape)
    np.testing.assert_array_equal(paddle.transpose(x, perm).numpy(),
                                  np.transpose(x.numpy(), perm))


@settings(max_examples=25, deadline=None)
@given(data=st.data())
def test_matmul_broadcast_semantics(data):
    b = data.draw(st.integers(1, 3))
    m = data.draw(st.integers(1, 5))
    k = data.draw(st.integers(1, 5))
    n = data.draw(st.integers(1, 5))
    x = paddle.randn([b, m, k])
    y = paddle.randn([k, n])
    out = paddle.matmul(x, y)
    np.testing.assert_allclose(out.numpy(), x.numpy() @ y.numpy(),
                               rtol=1e-4, atol=1e-5)
    xt = paddle.randn([b, k, m])
    out2 = paddle.matmul(xt, y, transpose_x=True)
    np.testing.assert_allclose(out2.numpy(),
                               np.swapaxes(xt.numpy(), -1, -2) @ y.numpy(),
                               rtol=1e-4, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(data=st.data())
def test_einsum_vs_numpy(data):
    i = data.draw(st.integers(1, 4))
    j = data.draw(st.integers(1, 4))
    k = data.draw(st.integers(1, 4))
    a = paddle.randn([i, j])
    b = paddle.randn([j, k])
    np.testing.assert_allclose(paddle.einsum("ij,jk->ik", a, b).numpy(),
                               np.einsum("ij,jk->ik", a.numpy(), b.numpy()),
                               rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(paddle.einsum("ij->j", a).numpy(),
                               a.numpy().sum(0), rtol=1e-4, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(shape=st.lists(st.integers(1, 5), min_size=2, max_size=2))
def test_linalg_matches_numpy(shape):
    n = max(shape)
    x = paddle.randn([n, n]) + paddle.eye(n) * n   # well-conditioned
    np.testing.assert_allclose(
        paddle.linalg.inv(x).numpy() @ x.numpy(), np.eye(n),
        rtol=1e-3, atol=1e-3)
    s = paddle.linalg.svd(x)[1]
    assert (s.numpy() >= -1e-6).all()


@settings(max_examples=20, deadline=None)
@given(data=st.data())
def test_indexing_semantics(data):
    n = data.draw(st.integers(2, 8))
    x = paddle.arange(n * 3, dtype="float32").reshape([n, 3])
    idx = data.draw(st.lists(st.integers(0, n - 1), min_size=1, max_size=4))
    it = paddle.to_tensor(idx, dtype="int64")
    np.testing.assert_array_equal(paddle.gather(x, it, axis=0).numpy(),
                                  x.numpy()[idx])
    np.testing.assert_array_equal(paddle.index_select(x, it, axis=0).numpy(),
                                  x.numpy()[idx])
