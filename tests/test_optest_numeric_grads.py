"""Reference-style OpTest numeric-gradient checks on the CPU oracle
implementations of the hot ops (SURVEY §4 test-strategy mirror)."""
import numpy as np
import torch

from op_test import OpTest

from paddle_amd.ops import functional as hot


class TestLayerNormOp(OpTest):
    rtol, atol = 1e-5, 1e-6

    def __init__(self):
        torch.manual_seed(0)

    def make_inputs(self):
        g = torch.Generator().manual_seed(3)
        return [torch.randn(3, 16, dtype=torch.float64, generator=g),
                torch.randn(16, dtype=torch.float64, generator=g),
                torch.randn(16, dtype=torch.float64, generator=g)]

    @staticmethod
    def fn(x, w, b):
        return hot.layer_norm(x.float(), w.float(), b.float(), 1e-5).double()

    @staticmethod
    def oracle(x, w, b):
        xn = x.numpy()
        mu = xn.mean(-1, keepdims=True)
        var = xn.var(-1, keepdims=True)
        return torch.from_numpy((xn - mu) / np.sqrt(var + 1e-5) * w.numpy() + b.numpy())


class TestRMSNormOp(OpTest):
    rtol, atol = 1e-5, 1e-6

    def make_inputs(self):
        g = torch.Generator().manual_seed(4)
        return [torch.randn(3, 16, dtype=torch.float64, generator=g),
                torch.randn(16, dtype=torch.float64, generator=g)]

    @staticmethod
    def fn(x, w):
        return hot.rms_norm(x.float(), w.float(), 1e-6).double()

    @staticmethod
    def oracle(x, w):
        xn = x.numpy()
        return torch.from_numpy(xn / np.sqrt((xn ** 2).mean(-1, keepdims=True) + 1e-6)
                                * w.numpy())


class TestSwigluOp(OpTest):
    rtol, atol = 1e-5, 1e-6

    def make_inputs(self):
        g = torch.Generator().manual_seed(5)
        return [torch.randn(4, 16, dtype=torch.float64, generator=g)]

    @staticmethod
    def fn(x):
        return hot.swiglu(x.float()).double()

    @staticmethod
    def oracle(x):
        xn = x.numpy()
        gg, u = xn[:, :8], xn[:, 8:]
        return torch.from_numpy(gg / (1 + np.exp(-gg)) * u)


def test_layer_norm_optest():
    t = TestLayerNormOp()
    t.check_output()
    t.check_grad()


def test_rms_norm_optest():
    t = TestRMSNormOp()
    t.check_output()
    t.check_grad()


def test_swiglu_optest():
    t = TestSwigluOp()
    t.check_output()
    t.check_grad()


class TestBiasGeluOp(OpTest):
    rtol, atol = 1e-4, 1e-5

    def make_inputs(self):
        g = torch.Generator().manual_seed(6)
        return [torch.randn(4, 16, dtype=torch.float64, generator=g),
                torch.randn(16, dtype=torch.float64, generator=g)]

    @staticmethod
    def fn(x, b):
        return hot.bias_gelu(x.float(), b.float()).double()

    @staticmethod
    def oracle(x, b):
        from scipy.special import erf
        v = x.numpy() + b.numpy()
        return torch.from_numpy(0.5 * v * (1 + erf(v / np.sqrt(2.0))))


class TestSoftmaxCrossEntropyOp(OpTest):
    rtol, atol = 1e-4, 1e-5
    grad_inputs = (0,)          # labels are integer; only logits get grads

    def make_inputs(self):
        g = torch.Generator().manual_seed(7)
        self.labels = torch.randint(0, 8, (5,), generator=g)
        return [torch.randn(5, 8, dtype=torch.float64, generator=g)]

    def fn(self, logits):
        return hot.softmax_cross_entropy(logits.float(), self.labels).double()

    def oracle(self, logits):
        ln = logits.numpy()
        lse = np.log(np.exp(ln - ln.max(-1, keepdims=True)).sum(-1)) + \
            ln.max(-1)
        picked = ln[np.arange(5), self.labels.numpy()]
        return torch.from_numpy(lse - picked)


class TestMaxoutOp(OpTest):
    rtol, atol = 1e-5, 1e-6

    def make_inputs(self):
        g = torch.Generator().manual_seed(8)
        return [torch.randn(2, 8, 3, dtype=torch.float64, generator=g)]

    @staticmethod
    def fn(x):
        import paddle_amd as paddle
        return paddle.nn.functional.maxout(x.float(), groups=2).double()

    @staticmethod
    def oracle(x):
        xn = x.numpy().reshape(2, 4, 2, 3)
        return torch.from_numpy(xn.max(axis=2))


def test_bias_gelu_optest():
    t = TestBiasGeluOp()
    t.check_output()
    t.check_grad()


def test_softmax_ce_optest():
    t = TestSoftmaxCrossEntropyOp()
    t.check_output()
    t.check_grad()


def test_maxout_optest():
    t = TestMaxoutOp()
    t.check_output()
    t.check_grad()


class TestRopeOp(OpTest):
    rtol, atol = 1e-4, 1e-5

    def make_inputs(self):
        g = torch.Generator().manual_seed(9)
        # [b=1, s=4, h=2, d=8]
        return [torch.randn(1, 4, 2, 8, dtype=torch.float64, generator=g)]

    def fn(self, x):
        from paddle_amd.ops.functional import _Rope, build_rope_cache
        cos, sin = build_rope_cache(4, 8, dtype=torch.float32)
        return _Rope.apply(x.float(), cos, sin, 0).double()

    def oracle(self, x):
        from paddle_amd.ops.functional import _rope_ref, build_rope_cache
        cos, sin = build_rope_cache(4, 8, dtype=torch.float64)
        return _rope_ref(x, cos, sin, 0, conj=False)


def test_rope_optest():
    t = TestRopeOp()
    t.check_output()
    t.check_grad()
