"""paddle.nn.functional parity (python/paddle/nn/functional/).

Hot ops dispatch to the gfx950 HIP kernels via paddle_amd.ops; the rest
map to torch.nn.functional with paddle conventions (weight layouts:
Linear weight is [in, out]).
"""
from __future__ import annotations

import math

import torch
import torch.nn.functional as TF

from ... import framework
from ...ops import functional as hot
from ...ops.functional import (  # noqa: F401
    flash_attention,
    scaled_dot_product_attention,
    fused_rotary_position_embedding,
    swiglu,
)

# re-export flash_attention submodule-style too
sdp_kernel = None


def linear(x, weight, bias=None, name=None):
    # paddle weight layout: [in_features, out_features].  Shapes the
    # autotune table assigns to the own MFMA kernel run gemm.hip's 8-phase
    # NT path (fwd via cached W^T, dgrad direct); everything else goes to
    # hipBLASLt via addmm (bias fused in its epilogue).
    if hot._own_linear_ok(x, weight):
        return hot.fused_linear_own(x, weight, bias)
    # NOTE: an M<=32 split-K weight-streaming kernel (decode_gemm.hip) was
    # measured SLOWER in-situ than hipBLASLt's skinny tiles (VALU-bound at
    # M=32); it stays available via _C.decode_gemm but is not dispatched.
    # Its MFMA rework (decode_gemm_mfma) IS dispatched, but only on the
    # tall-K decode shape where it beats hipBLASLt cold-HBM (fc2:
    # 51.3 vs 71.8 us at M32 K16384 N4096 -- tools/bench_decode_gemm.py);
    # lt already streams the wide-N shapes at 3.4-5.9 TB/s.
    if (not torch.is_grad_enabled() and x.dim() >= 2 and x.is_cuda
            and x.dtype == torch.bfloat16 and not hot._tracing()):
        mrows = x.numel() // x.shape[-1]
        k, n = weight.shape[-2], weight.shape[-1]
        if (mrows <= 32 and k >= 2 * n and k % 64 == 0 and n % 256 == 0
                and weight.stride(-1) == 1 and hot._ext.use_native(x)):
            C = hot._ext.get_ext()
            out = C.decode_gemm_mfma(x.reshape(-1, k), weight, bias)
            return out.reshape(*x.shape[:-1], n)
    if bias is not None and x.dim() >= 2:
        x2 = x.reshape(-1, x.shape[-1])
        out = torch.addmm(bias, x2, weight)
        return out.reshape(*x.shape[:-1], weight.shape[-1])
    out = torch.matmul(x, weight)
    if bias is not None:
        out = out + bias
    return out


def embedding(x, weight, padding_idx=None, sparse=False, name=None):
    return hot.embedding(x, weight, padding_idx)


def layer_norm(x, normalized_shape, weight=None, bias=None, epsilon=1e-5, name=None):
    if weight is None:
        return TF.layer_norm(x, normalized_shape if isinstance(normalized_shape, (list, tuple))
                             else (normalized_shape,), eps=epsilon)
    return hot.layer_norm(x, weight, bias, epsilon)


def rms_norm(x, weight, epsilon=1e-6):
    return hot.rms_norm(x, weight, epsilon)


def relu(x, name=None):
    return TF.relu(x)


def relu_(x):
    return TF.relu_(x)


def relu6(x, name=None):
    return TF.relu6(x)


def gelu(x, approximate=False, name=None):
    if x.is_cuda and not approximate:
        return hot.bias_gelu(x, None)
    return TF.gelu(x, approximate="tanh" if approximate else "none")


def silu(x, name=None):
    return TF.silu(x)


def sigmoid(x, name=None):
    return torch.sigmoid(x)


def tanh(x, name=None):
    return torch.tanh(x)


def softmax(x, axis=-1, dtype=None, name=None):
    dt = framework.convert_dtype(dtype) if dtype else None
    return TF.softmax(x, dim=axis, dtype=dt)


def log_softmax(x, axis=-1, dtype=None, name=None):
    dt = framework.convert_dtype(dtype) if dtype else None
    return TF.log_softmax(x, dim=axis, dtype=dt)


def leaky_relu(x, negative_slope=0.01, name=None):
    return TF.leaky_relu(x, negative_slope)


def elu(x, alpha=1.0, name=None):
    return TF.elu(x, alpha)


def hardswish(x, name=None):
    return TF.hardswish(x)


def hardsigmoid(x, slope=1 / 6, offset=0.5, name=None):
    return (x * slope + offset).clamp(0, 1)


def mish(x, name=None):
    return TF.mish(x)


def swish(x, name=None):
    return TF.silu(x)


def softplus(x, beta=1, threshold=20, name=None):
    return TF.softplus(x, beta, threshold)


def dropout(x, p=0.5, axis=None, training=True, mode="upscale_in_train", name=None):
    if not training or p == 0:
        return x
    if axis is not None:
        # paddle axis-dropout: mask broadcast along the other axes
        shape = [1] * x.dim()
        axes = [axis] if isinstance(axis, int) else list(axis)
        for a in axes:
            shape[a] = x.shape[a]
        mask = (torch.rand(shape, device=x.device) >= p).to(x.dtype)
        if mode == "upscale_in_train":
            return x * mask / (1 - p)
        return x * mask
    if mode == "upscale_in_train":
        return TF.dropout(x, p, training)
    return x * (torch.rand_like(x, dtype=torch.float32) >= p).to(x.dtype)


def dropout2d(x, p=0.5, training=True, data_format="NCHW", name=None):
    return TF.dropout2d(x, p, training)


def cross_entropy(input, label, weight=None, ignore_index=-100, reduction="mean",
                  soft_label=False, axis=-1, use_softmax=True, label_smoothing=0.0,
                  name=None):
    if soft_label or weight is not None or label_smoothing > 0 or not use_softmax:
        lf = input.float()
        return TF.cross_entropy(lf.reshape(-1, lf.shape[-1]),
                                label.reshape(-1) if not soft_label else label.reshape(-1, lf.shape[-1]),
                                weight=weight, ignore_index=ignore_index,
                                reduction=reduction, label_smoothing=label_smoothing)
    loss = hot.softmax_cross_entropy(input, label, ignore_index, reduction="none")
    if reduction == "mean":
        n_valid = (label != ignore_index).sum().clamp(min=1)
        return loss.sum() / n_valid.to(loss.dtype)
    if reduction == "sum":
        return loss.sum()
    return loss.unsqueeze(-1)  # paddle keeps trailing dim for reduction='none'


def softmax_with_cross_entropy(logits, label, soft_label=False, ignore_index=-100,
                               return_softmax=False, axis=-1):
    loss = hot.softmax_cross_entropy(logits, label.squeeze(-1) if label.dim() == logits.dim() else label,
                                     ignore_index, reduction="none").unsqueeze(-1)
    if return_softmax:
        return loss, TF.softmax(logits.float(), dim=axis).to(logits.dtype)
    return loss


def mse_loss(input, label, reduction="mean", name=None):
    return TF.mse_loss(input, label, reduction=reduction)


def l1_loss(input, label, reduction="mean", name=None):
    return TF.l1_loss(input, label, reduction=reduction)


def nll_loss(input, label, weight=None, ignore_index=-100, reduction="mean", name=None):
    return TF.nll_loss(input, label, weight, ignore_index=ignore_index, reduction=reduction)


def binary_cross_entropy(input, label, weight=None, reduction="mean", name=None):
    return TF.binary_cross_entropy(input, label, weight, reduction=reduction)


def binary_cross_entropy_with_logits(logit, label, weight=None, reduction="mean",
                                     pos_weight=None, name=None):
    return TF.binary_cross_entropy_with_logits(logit, label, weight, reduction=reduction,
                                               pos_weight=pos_weight)


def smooth_l1_loss(input, label, reduction="mean", delta=1.0, name=None):
    return TF.smooth_l1_loss(input, label, reduction=reduction, beta=delta)


def kl_div(input, label, reduction="mean", name=None):
    return TF.kl_div(input, label, reduction=reduction)


# -- conv / pool ------------------------------------------------------------
def _conv_padding(padding, x, weight, stride, dilation, nd):
    """paddle accepts padding "SAME"/"VALID" (any case, stride>1 ok);
    torch's "same" string rejects stride>1, so SAME computes explicit
    asymmetric padding (pads the high side like TF/paddle)."""
    if not isinstance(padding, str):
        return x, padding
    p = padding.lower()
    if p == "valid":
        return x, 0
    assert p == "same", padding
    strides = [stride] * nd if isinstance(stride, int) else list(stride)
    dils = [dilation] * nd if isinstance(dilation, int) else list(dilation)
    pads = []          # [lo, hi] per spatial dim, torch F.pad order reversed
    for i in range(nd):
        size = x.shape[2 + i]
        k = (weight.shape[2 + i] - 1) * dils[i] + 1
        out = -(-size // strides[i])
        total = max(0, (out - 1) * strides[i] + k - size)
        pads.append((total // 2, total - total // 2))
    if all(lo == hi for lo, hi in pads):
        return x, [lo for lo, _ in pads]
    flat = []
    for lo, hi in reversed(pads):
        flat += [lo, hi]
    return TF.pad(x, flat), 0


def conv2d(x, weight, bias=None, stride=1, padding=0, dilation=1, groups=1,
           data_format="NCHW", name=None):
    x, padding = _conv_padding(padding, x, weight, stride, dilation, 2)
    return TF.conv2d(x, weight, bias, stride, padding, dilation, groups)


def conv1d(x, weight, bias=None, stride=1, padding=0, dilation=1, groups=1,
           data_format="NCL", name=None):
    x, padding = _conv_padding(padding, x, weight, stride, dilation, 1)
    return TF.conv1d(x, weight, bias, stride, padding, dilation, groups)


def conv2d_transpose(x, weight, bias=None, stride=1, padding=0, output_padding=0,
                     groups=1, dilation=1, data_format="NCHW", output_size=None, name=None):
    return TF.conv_transpose2d(x, weight, bias, stride, padding, output_padding, groups, dilation)


def _pool_padding(padding, x, kernel_size, stride, nd, pad_value=0.0):
    """Pools also accept paddle's "SAME"/"VALID" strings (pad_value
    -inf for max pooling so padded cells never win)."""
    if not isinstance(padding, str):
        return x, padding
    p = padding.lower()
    if p == "valid":
        return x, 0
    assert p == "same", padding
    ks = [kernel_size] * nd if isinstance(kernel_size, int) else list(kernel_size)
    stride = stride or kernel_size
    strides = [stride] * nd if isinstance(stride, int) else list(stride)
    flat = []
    sym = []
    for i in range(nd):
        size = x.shape[2 + i]
        out = -(-size // strides[i])
        total = max(0, (out - 1) * strides[i] + ks[i] - size)
        sym.append((total // 2, total - total // 2))
    if all(lo == hi for lo, hi in sym):
        return x, [lo for lo, _ in sym]
    for lo, hi in reversed(sym):
        flat += [lo, hi]
    return TF.pad(x, flat, value=pad_value), 0


def max_pool2d(x, kernel_size, stride=None, padding=0, return_mask=False,
               ceil_mode=False, data_format="NCHW", name=None):
    x, padding = _pool_padding(padding, x, kernel_size, stride, 2,
                               pad_value=float("-inf"))
    out = TF.max_pool2d(x, kernel_size, stride, padding, ceil_mode=ceil_mode,
                        return_indices=return_mask)
    return out


def avg_pool2d(x, kernel_size, stride=None, padding=0, ceil_mode=False,
               exclusive=True, divisor_override=None, data_format="NCHW", name=None):
    x, padding = _pool_padding(padding, x, kernel_size, stride, 2)
    return TF.avg_pool2d(x, kernel_size, stride, padding, ceil_mode=ceil_mode,
                         count_include_pad=not exclusive, divisor_override=divisor_override)


def adaptive_avg_pool2d(x, output_size, data_format="NCHW", name=None):
    return TF.adaptive_avg_pool2d(x, output_size)


def batch_norm(x, running_mean, running_var, weight, bias, training=False,
               momentum=0.9, epsilon=1e-5, data_format="NCHW", use_global_stats=None,
               name=None):
    return TF.batch_norm(x, running_mean, running_var, weight, bias,
                         training=training, momentum=1 - momentum, eps=epsilon)


def interpolate(x, size=None, scale_factor=None, mode="nearest", align_corners=False,
                align_mode=0, data_format="NCHW", name=None):
    ac = align_corners if mode in ("linear", "bilinear", "bicubic", "trilinear") else None
    return TF.interpolate(x, size=size, scale_factor=scale_factor, mode=mode,
                          align_corners=ac)


def pad(x, pad, mode="constant", value=0.0, data_format="NCHW", name=None):
    return TF.pad(x, list(pad), mode=mode, value=value)


def unfold(x, kernel_sizes, strides=1, paddings=0, dilations=1, name=None):
    return TF.unfold(x, kernel_sizes, dilations, paddings, strides)


def one_hot(x, num_classes, name=None):
    return TF.one_hot(x.long(), num_classes).float()


def normalize(x, p=2, axis=1, epsilon=1e-12, name=None):
    return TF.normalize(x, p=p, dim=axis, eps=epsilon)


def glu(x, axis=-1, name=None):
    return TF.glu(x, dim=axis)


def label_smooth(label, prior_dist=None, epsilon=0.1, name=None):
    n = label.shape[-1]
    if prior_dist is not None:
        return (1 - epsilon) * label + epsilon * prior_dist
    return (1 - epsilon) * label + epsilon / n


# ---------------------------------------------------------------------------
# long-tail functional parity (reference: nn/functional/__init__.py __all__)
# -- thin torch.nn.functional dispatches + paddle-specific forms
# ---------------------------------------------------------------------------
import torch as _t
import torch.nn.functional as _F


def _drop_name(kw):
    kw.pop("name", None)
    return kw


def _mk(fn):
    def g(*a, **kw):
        return fn(*a, **_drop_name(kw))
    g.__name__ = fn.__name__
    return g


celu = _mk(_F.celu)
hardshrink = _mk(_F.hardshrink)
log_sigmoid = _mk(_F.logsigmoid)
prelu = _mk(_F.prelu)
selu = _mk(_F.selu)
softshrink = _mk(_F.softshrink)
softsign = _mk(_F.softsign)
tanhshrink = _mk(_F.tanhshrink)
gumbel_softmax = _mk(_F.gumbel_softmax)
alpha_dropout = _mk(_F.alpha_dropout)
feature_alpha_dropout = _mk(_F.feature_alpha_dropout)
avg_pool1d = _mk(_F.avg_pool1d)
avg_pool3d = _mk(_F.avg_pool3d)
lp_pool1d = _mk(_F.lp_pool1d)
lp_pool2d = _mk(_F.lp_pool2d)
max_pool1d = _mk(_F.max_pool1d)
max_pool3d = _mk(_F.max_pool3d)
max_unpool1d = _mk(_F.max_unpool1d)
max_unpool2d = _mk(_F.max_unpool2d)
max_unpool3d = _mk(_F.max_unpool3d)
adaptive_avg_pool1d = _mk(_F.adaptive_avg_pool1d)
adaptive_avg_pool3d = _mk(_F.adaptive_avg_pool3d)
adaptive_max_pool1d = _mk(_F.adaptive_max_pool1d)
adaptive_max_pool2d = _mk(_F.adaptive_max_pool2d)
adaptive_max_pool3d = _mk(_F.adaptive_max_pool3d)
fractional_max_pool2d = _mk(_F.fractional_max_pool2d)
fractional_max_pool3d = _mk(_F.fractional_max_pool3d)
margin_ranking_loss = _mk(_F.margin_ranking_loss)
multi_label_soft_margin_loss = _mk(_F.multilabel_soft_margin_loss)
poisson_nll_loss = _mk(_F.poisson_nll_loss)
ctc_loss = _mk(_F.ctc_loss)
hinge_embedding_loss = _mk(_F.hinge_embedding_loss)
affine_grid = _mk(_F.affine_grid)
grid_sample = _mk(_F.grid_sample)
local_response_norm = _mk(_F.local_response_norm)
pixel_shuffle = _mk(_F.pixel_shuffle)
pixel_unshuffle = _mk(_F.pixel_unshuffle)
channel_shuffle = _mk(_F.channel_shuffle)
instance_norm = _mk(_F.instance_norm)
group_norm = _mk(_F.group_norm)
fold = _mk(_F.fold)
cosine_similarity = _mk(_F.cosine_similarity)
cosine_embedding_loss = _mk(_F.cosine_embedding_loss)
rrelu = _mk(_F.rrelu)
triplet_margin_loss = _mk(_F.triplet_margin_loss)
triplet_margin_with_distance_loss = _mk(_F.triplet_margin_with_distance_loss)
soft_margin_loss = _mk(_F.soft_margin_loss)
gaussian_nll_loss = _mk(_F.gaussian_nll_loss)
multi_margin_loss = _mk(_F.multi_margin_loss)
pairwise_distance = _mk(_F.pairwise_distance)
bilinear = _mk(_F.bilinear)
dropout3d = _mk(_F.dropout3d)
conv3d = _mk(_F.conv3d)
conv3d_transpose = _mk(_F.conv_transpose3d)
conv1d_transpose = _mk(_F.conv_transpose1d)
hardtanh = _mk(_F.hardtanh)


def upsample(x, size=None, scale_factor=None, mode="nearest",
             align_corners=False, align_mode=0, data_format="NCHW", name=None):
    ac = align_corners if mode in ("linear", "bilinear", "bicubic",
                                   "trilinear") else None
    return _F.interpolate(x, size=size, scale_factor=scale_factor, mode=mode,
                          align_corners=ac)


def zeropad2d(x, padding, data_format="NCHW", name=None):
    return _F.pad(x, padding if isinstance(padding, (list, tuple)) else
                  [padding] * 4)


def maxout(x, groups, axis=1, name=None):
    shape = list(x.shape)
    c = shape[axis]
    assert c % groups == 0
    new = shape[:axis] + [c // groups, groups] + shape[axis + 1:]
    return x.reshape(new).max(dim=axis + 1).values


def thresholded_relu(x, threshold=1.0, value=0.0, name=None):
    return _t.where(x > threshold, x, _t.full_like(x, value))


def sequence_mask(x, maxlen=None, dtype="int64", name=None):
    from ... import framework as _fw
    m = int(maxlen) if maxlen is not None else int(x.max())
    r = _t.arange(m, device=x.device)
    return (r.unsqueeze(0) < x.unsqueeze(-1)).to(_fw.convert_dtype(dtype))


def dice_loss(input, label, epsilon=1e-5, name=None):
    label_one_hot = _F.one_hot(label.squeeze(-1).long(), input.shape[-1]).to(input.dtype)
    inter = (input * label_one_hot).sum(-1)
    union = input.sum(-1) + label_one_hot.sum(-1)
    return (1 - (2 * inter + epsilon) / (union + epsilon)).mean()


def log_loss(input, label, epsilon=1e-4, name=None):
    return -label * _t.log(input + epsilon) - (1 - label) * _t.log(
        1 - input + epsilon)


def square_error_cost(input, label):
    return (input - label) ** 2


def npair_loss(anchor, positive, labels, l2_reg=0.002):
    sim = anchor @ positive.t()
    ce = _F.cross_entropy(sim, _t.arange(sim.shape[0], device=sim.device))
    reg = l2_reg * (anchor.pow(2).sum(1).mean() + positive.pow(2).sum(1).mean())
    return ce + reg


def sigmoid_focal_loss(logit, label, normalizer=None, alpha=0.25, gamma=2.0,
                       reduction="sum", name=None):
    p = _t.sigmoid(logit)
    ce = _F.binary_cross_entropy_with_logits(logit, label, reduction="none")
    pt = p * label + (1 - p) * (1 - label)
    af = alpha * label + (1 - alpha) * (1 - label)
    loss = af * (1 - pt) ** gamma * ce
    if normalizer is not None:
        loss = loss / normalizer
    if reduction == "sum":
        return loss.sum()
    if reduction == "mean":
        return loss.mean()
    return loss


def margin_cross_entropy(logits, label, margin1=1.0, margin2=0.5, margin3=0.0,
                         scale=64.0, group=None, return_softmax=False,
                         reduction="mean"):
    """ArcFace/CosFace-style margin softmax (single-rank form; the
    model-parallel variant lives in fleet.mpu.ParallelCrossEntropy).
    reference: paddle/fluid/operators/margin_cross_entropy_op.cu"""
    cos = logits.float().clamp(-1, 1)
    theta = _t.acos(cos.gather(1, label.view(-1, 1)))
    target = _t.cos(margin1 * theta + margin2) - margin3
    out = cos.scatter(1, label.view(-1, 1), target) * scale
    loss = _F.cross_entropy(out, label.view(-1), reduction=reduction)
    if return_softmax:
        return loss, _F.softmax(out, -1)
    return loss


def temporal_shift(x, seg_num, shift_ratio=0.25, data_format="NCHW", name=None):
    nt, c, h, w = x.shape
    n = nt // seg_num
    x5 = x.view(n, seg_num, c, h, w)
    fold_c = int(c * shift_ratio)
    out = _t.zeros_like(x5)
    out[:, 1:, :fold_c] = x5[:, :-1, :fold_c]              # shift left
    out[:, :-1, fold_c:2 * fold_c] = x5[:, 1:, fold_c:2 * fold_c]  # right
    out[:, :, 2 * fold_c:] = x5[:, :, 2 * fold_c:]
    return out.view(nt, c, h, w)


def gather_tree(ids, parents):
    max_len, batch, beam = ids.shape
    out = _t.zeros_like(ids)
    out[-1] = ids[-1]
    parent = parents[-1]
    for t in range(max_len - 2, -1, -1):
        b_idx = _t.arange(batch, device=ids.device).unsqueeze(1).expand(batch, beam)
        out[t] = ids[t][b_idx, parent]
        parent = parents[t][b_idx, parent]
    return out


def flash_attn_qkvpacked(qkv, dropout=0.0, causal=False, return_softmax=False,
                         **kwargs):
    """Packed [B,S,3,H,D] flash attention on the zero-copy HIP path."""
    from ...ops import functional as hot
    return hot.qkv_flash_attention(qkv, causal=causal), None


def _gated(name, why):
    def f(*a, **k):
        raise NotImplementedError(f"{name}: {why}")
    f.__name__ = name
    return f


hsigmoid_loss = _gated("hsigmoid_loss", "hierarchical sigmoid not in this build")
rnnt_loss = _gated("rnnt_loss", "transducer kernel not in this build")
class_center_sample = _gated("class_center_sample", "PLSC-style sampling: round 2")
sparse_attention = _gated("sparse_attention", "use flash_attention (dense) or paged decode")
adaptive_log_softmax_with_loss = _gated("adaptive_log_softmax_with_loss",
                                        "use nn.AdaptiveLogSoftmaxWithLoss layer")
flashmask_attention = _gated("flashmask_attention", "mask-sparse FA: round 2")
def flash_attn_varlen_qkvpacked(qkv, cu_seqlens, max_seqlen, dropout=0.0,
                                causal=False, **kwargs):
    """Packed ragged QKV [total, 3, H, D] -> varlen flash attention."""
    from ...ops.functional import flash_attn_varlen_func
    q, k, v = qkv.unbind(1)
    return flash_attn_varlen_func(q, k, v, cu_seqlens, cu_seqlens,
                                  max_seqlen, max_seqlen, causal=causal), None


def _mk_inplace(fn, extract=None):
    def g(x, *a, **kw):
        kw.pop("name", None)
        out = fn(x, *a, **kw)
        x.copy_(out)
        return x
    return g


elu_ = _mk_inplace(_F.elu)
hardtanh_ = _mk_inplace(_F.hardtanh)
leaky_relu_ = _mk_inplace(_F.leaky_relu)
softmax_ = _mk_inplace(_F.softmax)
tanh_ = _mk_inplace(_t.tanh)
thresholded_relu_ = _mk_inplace(thresholded_relu)
