"""End-to-end training sanity: LeNet (BASELINE config 1) + tiny GPT loss
descent; sharding stage-3 parity at world=1."""
import copy

import numpy as np
import pytest
import torch

import paddle_amd as paddle
from paddle_amd import nn
from paddle_amd.distributed.fleet.sharding import GroupShardedStage3, ShardedAdamW
from paddle_amd.models import GPTPretrainingCriterion, build_gpt
from paddle_amd.vision.models import LeNet


def test_lenet_mnist_dygraph_cpu():
    """BASELINE config 1: MNIST LeNet dygraph on CPUPlace (synthetic data)."""
    paddle.seed(42)
    model = LeNet(num_classes=10)
    opt = paddle.optimizer.Adam(learning_rate=1e-3, parameters=model.parameters())
    loss_fn = nn.CrossEntropyLoss()
    x = paddle.randn([32, 1, 28, 28])
    y = paddle.randint(0, 10, (32,))
    losses = []
    for _ in range(20):
        loss = loss_fn(model(x), y)
        loss.backward()
        opt.step()
        opt.clear_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.5, losses


def test_gpt_tiny_loss_decreases():
    paddle.seed(0)
    m = build_gpt("gpt3-tiny")
    loss_fn = GPTPretrainingCriterion()
    opt = paddle.optimizer.AdamW(learning_rate=3e-4, parameters=m.parameters())
    ids = paddle.randint(0, 1024, (2, 64))
    losses = []
    for _ in range(10):
        loss = loss_fn(m(ids), ids)
        loss.backward()
        opt.step()
        opt.clear_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0] - 0.5, losses


def test_sharding3_world1_matches_plain():
    paddle.seed(0)
    m = build_gpt("gpt3-tiny")
    loss_fn = GPTPretrainingCriterion()
    ids = paddle.randint(0, 1024, (2, 64))
    m2 = copy.deepcopy(m)
    opt2 = paddle.optimizer.AdamW(learning_rate=1e-3, parameters=m2.parameters(),
                                  weight_decay=0.01)
    for _ in range(3):
        l2 = loss_fn(m2(ids), ids)
        l2.backward()
        opt2.step()
        opt2.clear_grad()
    wrapped = GroupShardedStage3(m)
    opt = ShardedAdamW(wrapped, learning_rate=1e-3, weight_decay=0.01)
    for _ in range(3):
        l1 = loss_fn(wrapped(ids), ids)
        l1.backward()
        opt.step()
        opt.clear_grad()
    assert abs(float(l1) - float(l2)) < 1e-4


def test_sharding3_state_dict_roundtrip():
    paddle.seed(0)
    m = build_gpt("gpt3-tiny")
    wrapped = GroupShardedStage3(m)
    sd = wrapped.state_dict()
    assert "gpt.embeddings.word_embeddings.weight" in sd
    wrapped2 = GroupShardedStage3(build_gpt("gpt3-tiny"))
    wrapped2.set_state_dict(sd)
    ids = paddle.randint(0, 1024, (1, 32))
    with torch.no_grad():
        o1 = wrapped(ids)
        o2 = wrapped2(ids)
    torch.testing.assert_close(o1, o2)


def test_gpt_recompute_matches():
    paddle.seed(0)
    m = build_gpt("gpt3-tiny")
    m_rc = copy.deepcopy(m)
    m_rc.cfg.use_recompute = True
    m_rc.gpt.cfg.use_recompute = True
    loss_fn = GPTPretrainingCriterion()
    ids = paddle.randint(0, 1024, (2, 32))
    l1 = loss_fn(m(ids), ids)
    l1.backward()
    l2 = loss_fn(m_rc(ids), ids)
    l2.backward()
    torch.testing.assert_close(l1, l2, rtol=1e-5, atol=1e-6)
    for p1, p2 in zip(m.parameters(), m_rc.parameters()):
        torch.testing.assert_close(p1.grad, p2.grad, rtol=1e-4, atol=1e-5)


def test_hapi_style_eval_mode():
    m = build_gpt("gpt3-tiny")
    m.eval()
    ids = paddle.randint(0, 1024, (1, 16))
    with paddle.no_grad():
        out = m(ids)
    assert out.shape == (1, 16, 1024)
