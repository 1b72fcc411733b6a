"""Tensor API vs numpy oracle (reference test style: test/legacy_test
OpTest numpy comparisons)."""
import numpy as np
import pytest
import torch

import paddle_amd as paddle


def test_creation():
    assert paddle.zeros([2, 3]).shape == (2, 3)
    assert paddle.ones([2], dtype="int64").dtype == torch.int64
    t = paddle.to_tensor(np.arange(6).reshape(2, 3).astype("float32"))
    np.testing.assert_allclose(t.numpy(), np.arange(6).reshape(2, 3))
    assert paddle.full([2, 2], 7).numpy().tolist() == [[7, 7], [7, 7]]
    assert paddle.arange(5).numpy().tolist() == [0, 1, 2, 3, 4]
    e = paddle.eye(3)
    np.testing.assert_allclose(e.numpy(), np.eye(3))


def test_stop_gradient_property():
    t = paddle.to_tensor([1.0, 2.0])
    assert t.stop_gradient
    t.stop_gradient = False
    assert t.requires_grad
    t.stop_gradient = True
    assert not t.requires_grad


def test_manipulation():
    x = paddle.to_tensor(np.arange(24).reshape(2, 3, 4).astype("float32"))
    assert paddle.reshape(x, [6, 4]).shape == (6, 4)
    assert paddle.transpose(x, [2, 0, 1]).shape == (4, 2, 3)
    assert paddle.concat([x, x], axis=0).shape == (4, 3, 4)
    parts = paddle.split(x, 3, axis=1)
    assert len(parts) == 3 and parts[0].shape == (2, 1, 4)
    parts = paddle.split(x, [1, 2], axis=1)
    assert parts[1].shape == (2, 2, 4)
    assert paddle.squeeze(paddle.unsqueeze(x, 0), 0).shape == x.shape
    assert paddle.flatten(x, 1).shape == (2, 12)
    assert paddle.tile(paddle.ones([2]), [3]).shape == (6,)
    st = paddle.stack([x, x], axis=0)
    assert st.shape == (2, 2, 3, 4)


def test_math_reductions():
    a = np.random.rand(3, 4).astype("float32")
    x = paddle.to_tensor(a)
    np.testing.assert_allclose(paddle.sum(x).numpy(), a.sum(), rtol=1e-6)
    np.testing.assert_allclose(paddle.sum(x, axis=1).numpy(), a.sum(1), rtol=1e-6)
    np.testing.assert_allclose(paddle.mean(x, axis=0, keepdim=True).numpy(),
                               a.mean(0, keepdims=True), rtol=1e-6)
    np.testing.assert_allclose(paddle.max(x, axis=1).numpy(), a.max(1), rtol=1e-6)
    np.testing.assert_allclose(paddle.logsumexp(x, axis=-1).numpy(),
                               np.log(np.exp(a).sum(-1)), rtol=1e-5)
    np.testing.assert_allclose(paddle.cumsum(x, axis=0).numpy(), a.cumsum(0), rtol=1e-6)


def test_matmul_transpose_args():
    a = np.random.rand(3, 4).astype("float32")
    b = np.random.rand(5, 4).astype("float32")
    out = paddle.matmul(paddle.to_tensor(a), paddle.to_tensor(b), transpose_y=True)
    np.testing.assert_allclose(out.numpy(), a @ b.T, rtol=1e-5)


def test_gather_scatter_index():
    x = paddle.to_tensor(np.arange(12).reshape(4, 3).astype("float32"))
    idx = paddle.to_tensor([0, 2])
    assert paddle.gather(x, idx).numpy().tolist() == [[0, 1, 2], [6, 7, 8]]
    y = paddle.index_select(x, idx, axis=1)
    assert y.shape == (4, 2)
    w = paddle.where(x > 5, paddle.ones_like(x), paddle.zeros_like(x))
    assert w.numpy().sum() == 6


def test_comparison_and_logic():
    x = paddle.to_tensor([1.0, 2.0, 3.0])
    y = paddle.to_tensor([3.0, 2.0, 1.0])
    assert paddle.equal(x, y).numpy().tolist() == [False, True, False]
    assert bool(paddle.allclose(x, x))
    assert paddle.logical_and(x > 1, y > 1).numpy().tolist() == [False, True, False]


def test_einsum_topk_sort():
    a = np.random.rand(2, 3).astype("float32")
    b = np.random.rand(3, 4).astype("float32")
    out = paddle.einsum("ij,jk->ik", paddle.to_tensor(a), paddle.to_tensor(b))
    np.testing.assert_allclose(out.numpy(), a @ b, rtol=1e-5)
    v, i = paddle.topk(paddle.to_tensor([1.0, 3.0, 2.0]), 2)
    assert v.numpy().tolist() == [3.0, 2.0]
    assert paddle.argsort(paddle.to_tensor([3.0, 1.0, 2.0])).numpy().tolist() == [1, 2, 0]


def test_cast_astype():
    x = paddle.ones([2], dtype="float32")
    assert x.astype("int64").dtype == torch.int64
    assert paddle.cast(x, "float16").dtype == torch.float16


def test_reshape_zero_copies_dim():
    """paddle reshape semantics: 0 copies the corresponding input dim
    (reference tensor/manipulation.py reshape); -1 still infers."""
    import paddle_amd as paddle
    x = paddle.ones([2, 3, 4])
    assert tuple(paddle.reshape(x, [0, 3, 4]).shape) == (2, 3, 4)
    assert tuple(paddle.reshape(x, [0, -1]).shape) == (2, 12)
    assert tuple(paddle.reshape(x, [0, 0, -1]).shape) == (2, 3, 4)
