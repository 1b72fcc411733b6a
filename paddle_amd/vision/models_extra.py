"""Long-tail vision model families (reference: python/paddle/vision/models/
{vgg,alexnet,squeezenet,densenet,inceptionv3,googlenet,mobilenetv1,
mobilenetv2,mobilenetv3,shufflenetv2}.py).

Compact re-derivations on the torch substrate (conv/bn/pool run on
MIOpen).  `pretrained=True` is unavailable (no network egress) and
raises.
"""
from __future__ import annotations

import torch
import torch.nn as tn

from ..nn.layer import Layer


def _no_pretrained(pretrained):
    if pretrained:
        raise RuntimeError("pretrained weights need network egress; "
                           "load a local checkpoint with paddle.load instead")


# ---------------------------------------------------------------------------
# VGG
# ---------------------------------------------------------------------------
_VGG_CFGS = {
    "A": [64, "M", 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "B": [64, 64, "M", 128, 128, "M", 256, 256, "M", 512, 512, "M", 512, 512, "M"],
    "D": [64, 64, "M", 128, 128, "M", 256, 256, 256, "M", 512, 512, 512, "M",
          512, 512, 512, "M"],
    "E": [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M", 512, 512, 512,
          512, "M", 512, 512, 512, 512, "M"],
}


class VGG(Layer):
    def __init__(self, features, num_classes=1000, with_pool=True):
        super().__init__()
        self.features = features
        self.with_pool = with_pool
        if with_pool:
            self.avgpool = tn.AdaptiveAvgPool2d((7, 7))
        self.classifier = tn.Sequential(
            tn.Linear(512 * 7 * 7, 4096), tn.ReLU(True), tn.Dropout(),
            tn.Linear(4096, 4096), tn.ReLU(True), tn.Dropout(),
            tn.Linear(4096, num_classes))

    def forward(self, x):
        x = self.features(x)
        if self.with_pool:
            x = self.avgpool(x)
        return self.classifier(torch.flatten(x, 1))


def _vgg_features(cfg, batch_norm=False):
    layers, c = [], 3
    for v in _VGG_CFGS[cfg]:
        if v == "M":
            layers.append(tn.MaxPool2d(2, 2))
        else:
            layers.append(tn.Conv2d(c, v, 3, padding=1))
            if batch_norm:
                layers.append(tn.BatchNorm2d(v))
            layers.append(tn.ReLU(True))
            c = v
    return tn.Sequential(*layers)


def vgg11(pretrained=False, batch_norm=False, **kw):
    _no_pretrained(pretrained)
    return VGG(_vgg_features("A", batch_norm), **kw)


def vgg13(pretrained=False, batch_norm=False, **kw):
    _no_pretrained(pretrained)
    return VGG(_vgg_features("B", batch_norm), **kw)


def vgg16(pretrained=False, batch_norm=False, **kw):
    _no_pretrained(pretrained)
    return VGG(_vgg_features("D", batch_norm), **kw)


def vgg19(pretrained=False, batch_norm=False, **kw):
    _no_pretrained(pretrained)
    return VGG(_vgg_features("E", batch_norm), **kw)


# ---------------------------------------------------------------------------
# AlexNet / SqueezeNet
# ---------------------------------------------------------------------------
class AlexNet(Layer):
    def __init__(self, num_classes=1000):
        super().__init__()
        self.features = tn.Sequential(
            tn.Conv2d(3, 64, 11, 4, 2), tn.ReLU(True), tn.MaxPool2d(3, 2),
            tn.Conv2d(64, 192, 5, padding=2), tn.ReLU(True), tn.MaxPool2d(3, 2),
            tn.Conv2d(192, 384, 3, padding=1), tn.ReLU(True),
            tn.Conv2d(384, 256, 3, padding=1), tn.ReLU(True),
            tn.Conv2d(256, 256, 3, padding=1), tn.ReLU(True), tn.MaxPool2d(3, 2))
        self.avgpool = tn.AdaptiveAvgPool2d((6, 6))
        self.classifier = tn.Sequential(
            tn.Dropout(), tn.Linear(256 * 36, 4096), tn.ReLU(True),
            tn.Dropout(), tn.Linear(4096, 4096), tn.ReLU(True),
            tn.Linear(4096, num_classes))

    def forward(self, x):
        return self.classifier(torch.flatten(self.avgpool(self.features(x)), 1))


def alexnet(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return AlexNet(**kw)


class _Fire(Layer):
    def __init__(self, cin, squeeze, e1, e3):
        super().__init__()
        self.squeeze = tn.Conv2d(cin, squeeze, 1)
        self.e1 = tn.Conv2d(squeeze, e1, 1)
        self.e3 = tn.Conv2d(squeeze, e3, 3, padding=1)

    def forward(self, x):
        x = torch.relu(self.squeeze(x))
        return torch.cat([torch.relu(self.e1(x)), torch.relu(self.e3(x))], 1)


class SqueezeNet(Layer):
    def __init__(self, version="1.0", num_classes=1000):
        super().__init__()
        if version == "1.0":
            self.features = tn.Sequential(
                tn.Conv2d(3, 96, 7, 2), tn.ReLU(True), tn.MaxPool2d(3, 2, ceil_mode=True),
                _Fire(96, 16, 64, 64), _Fire(128, 16, 64, 64), _Fire(128, 32, 128, 128),
                tn.MaxPool2d(3, 2, ceil_mode=True),
                _Fire(256, 32, 128, 128), _Fire(256, 48, 192, 192),
                _Fire(384, 48, 192, 192), _Fire(384, 64, 256, 256),
                tn.MaxPool2d(3, 2, ceil_mode=True), _Fire(512, 64, 256, 256))
        else:
            self.features = tn.Sequential(
                tn.Conv2d(3, 64, 3, 2), tn.ReLU(True), tn.MaxPool2d(3, 2, ceil_mode=True),
                _Fire(64, 16, 64, 64), _Fire(128, 16, 64, 64),
                tn.MaxPool2d(3, 2, ceil_mode=True),
                _Fire(128, 32, 128, 128), _Fire(256, 32, 128, 128),
                tn.MaxPool2d(3, 2, ceil_mode=True),
                _Fire(256, 48, 192, 192), _Fire(384, 48, 192, 192),
                _Fire(384, 64, 256, 256), _Fire(512, 64, 256, 256))
        self.classifier = tn.Sequential(
            tn.Dropout(), tn.Conv2d(512, num_classes, 1), tn.ReLU(True),
            tn.AdaptiveAvgPool2d((1, 1)))

    def forward(self, x):
        return torch.flatten(self.classifier(self.features(x)), 1)


def squeezenet1_0(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return SqueezeNet("1.0", **kw)


def squeezenet1_1(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return SqueezeNet("1.1", **kw)


# ---------------------------------------------------------------------------
# DenseNet
# ---------------------------------------------------------------------------
class _DenseLayer(Layer):
    def __init__(self, cin, growth, bn_size):
        super().__init__()
        self.net = tn.Sequential(
            tn.BatchNorm2d(cin), tn.ReLU(True),
            tn.Conv2d(cin, bn_size * growth, 1, bias=False),
            tn.BatchNorm2d(bn_size * growth), tn.ReLU(True),
            tn.Conv2d(bn_size * growth, growth, 3, padding=1, bias=False))

    def forward(self, x):
        return torch.cat([x, self.net(x)], 1)


_DENSE_CFGS = {121: (32, [6, 12, 24, 16]), 161: (48, [6, 12, 36, 24]),
               169: (32, [6, 12, 32, 32]), 201: (32, [6, 12, 48, 32]),
               264: (32, [6, 12, 64, 48])}


class DenseNet(Layer):
    def __init__(self, layers=121, bn_size=4, dropout=0.0, num_classes=1000,
                 with_pool=True):
        super().__init__()
        growth, block_cfg = _DENSE_CFGS[layers]
        c = 2 * growth
        feats = [tn.Conv2d(3, c, 7, 2, 3, bias=False), tn.BatchNorm2d(c),
                 tn.ReLU(True), tn.MaxPool2d(3, 2, 1)]
        for i, n in enumerate(block_cfg):
            for _ in range(n):
                feats.append(_DenseLayer(c, growth, bn_size))
                c += growth
            if i != len(block_cfg) - 1:
                feats += [tn.BatchNorm2d(c), tn.ReLU(True),
                          tn.Conv2d(c, c // 2, 1, bias=False), tn.AvgPool2d(2, 2)]
                c //= 2
        feats += [tn.BatchNorm2d(c), tn.ReLU(True)]
        self.features = tn.Sequential(*feats)
        self.with_pool = with_pool
        self.fc = tn.Linear(c, num_classes)

    def forward(self, x):
        x = self.features(x)
        if self.with_pool:
            x = torch.nn.functional.adaptive_avg_pool2d(x, 1)
        return self.fc(torch.flatten(x, 1))


def densenet121(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return DenseNet(121, **kw)


def densenet161(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return DenseNet(161, **kw)


def densenet169(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return DenseNet(169, **kw)


def densenet201(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return DenseNet(201, **kw)


def densenet264(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return DenseNet(264, **kw)


# ---------------------------------------------------------------------------
# MobileNet V1/V2/V3
# ---------------------------------------------------------------------------
def _cbr(cin, cout, k=3, s=1, g=1, act=tn.ReLU6):
    return tn.Sequential(
        tn.Conv2d(cin, cout, k, s, k // 2, groups=g, bias=False),
        tn.BatchNorm2d(cout), act(inplace=True))


class MobileNetV1(Layer):
    def __init__(self, scale=1.0, num_classes=1000, with_pool=True):
        super().__init__()
        def c(x):
            return max(8, int(x * scale))
        cfg = [(32, 64, 1), (64, 128, 2), (128, 128, 1), (128, 256, 2),
               (256, 256, 1), (256, 512, 2)] + [(512, 512, 1)] * 5 + \
              [(512, 1024, 2), (1024, 1024, 1)]
        layers = [_cbr(3, c(32), 3, 2, act=tn.ReLU)]
        for cin, cout, s in cfg:
            layers += [_cbr(c(cin), c(cin), 3, s, g=c(cin), act=tn.ReLU),
                       _cbr(c(cin), c(cout), 1, 1, act=tn.ReLU)]
        self.features = tn.Sequential(*layers)
        self.fc = tn.Linear(c(1024), num_classes)

    def forward(self, x):
        x = torch.nn.functional.adaptive_avg_pool2d(self.features(x), 1)
        return self.fc(torch.flatten(x, 1))


def mobilenet_v1(pretrained=False, scale=1.0, **kw):
    _no_pretrained(pretrained)
    return MobileNetV1(scale, **kw)


class _InvertedResidual(Layer):
    def __init__(self, cin, cout, stride, expand):
        super().__init__()
        hidden = int(round(cin * expand))
        self.use_res = stride == 1 and cin == cout
        layers = []
        if expand != 1:
            layers.append(_cbr(cin, hidden, 1))
        layers += [_cbr(hidden, hidden, 3, stride, g=hidden),
                   tn.Conv2d(hidden, cout, 1, bias=False), tn.BatchNorm2d(cout)]
        self.conv = tn.Sequential(*layers)

    def forward(self, x):
        return x + self.conv(x) if self.use_res else self.conv(x)


class MobileNetV2(Layer):
    def __init__(self, scale=1.0, num_classes=1000, with_pool=True):
        super().__init__()
        cfg = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
               (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]
        def c(x):
            return max(8, int(x * scale))
        layers = [_cbr(3, c(32), 3, 2)]
        cin = c(32)
        for t, ch, n, s in cfg:
            for i in range(n):
                layers.append(_InvertedResidual(cin, c(ch), s if i == 0 else 1, t))
                cin = c(ch)
        layers.append(_cbr(cin, c(1280), 1))
        self.features = tn.Sequential(*layers)
        self.classifier = tn.Sequential(tn.Dropout(0.2),
                                        tn.Linear(c(1280), num_classes))

    def forward(self, x):
        x = torch.nn.functional.adaptive_avg_pool2d(self.features(x), 1)
        return self.classifier(torch.flatten(x, 1))


def mobilenet_v2(pretrained=False, scale=1.0, **kw):
    _no_pretrained(pretrained)
    return MobileNetV2(scale, **kw)


class _SE(Layer):
    def __init__(self, c, r=4):
        super().__init__()
        self.fc = tn.Sequential(tn.Linear(c, c // r), tn.ReLU(True),
                                tn.Linear(c // r, c), tn.Hardsigmoid())

    def forward(self, x):
        s = torch.nn.functional.adaptive_avg_pool2d(x, 1).flatten(1)
        return x * self.fc(s).unsqueeze(-1).unsqueeze(-1)


class _V3Block(Layer):
    def __init__(self, cin, hidden, cout, k, stride, se, act):
        super().__init__()
        A = tn.Hardswish if act == "HS" else tn.ReLU
        self.use_res = stride == 1 and cin == cout
        layers = []
        if hidden != cin:
            layers.append(_cbr(cin, hidden, 1, act=A))
        layers.append(_cbr(hidden, hidden, k, stride, g=hidden, act=A))
        if se:
            layers.append(_SE(hidden))
        layers += [tn.Conv2d(hidden, cout, 1, bias=False), tn.BatchNorm2d(cout)]
        self.conv = tn.Sequential(*layers)

    def forward(self, x):
        return x + self.conv(x) if self.use_res else self.conv(x)


_V3_SMALL = [(16, 16, 16, 3, 2, True, "RE"), (16, 72, 24, 3, 2, False, "RE"),
             (24, 88, 24, 3, 1, False, "RE"), (24, 96, 40, 5, 2, True, "HS"),
             (40, 240, 40, 5, 1, True, "HS"), (40, 240, 40, 5, 1, True, "HS"),
             (40, 120, 48, 5, 1, True, "HS"), (48, 144, 48, 5, 1, True, "HS"),
             (48, 288, 96, 5, 2, True, "HS"), (96, 576, 96, 5, 1, True, "HS"),
             (96, 576, 96, 5, 1, True, "HS")]
_V3_LARGE = [(16, 16, 16, 3, 1, False, "RE"), (16, 64, 24, 3, 2, False, "RE"),
             (24, 72, 24, 3, 1, False, "RE"), (24, 72, 40, 5, 2, True, "RE"),
             (40, 120, 40, 5, 1, True, "RE"), (40, 120, 40, 5, 1, True, "RE"),
             (40, 240, 80, 3, 2, False, "HS"), (80, 200, 80, 3, 1, False, "HS"),
             (80, 184, 80, 3, 1, False, "HS"), (80, 184, 80, 3, 1, False, "HS"),
             (80, 480, 112, 3, 1, True, "HS"), (112, 672, 112, 3, 1, True, "HS"),
             (112, 672, 160, 5, 2, True, "HS"), (160, 960, 160, 5, 1, True, "HS"),
             (160, 960, 160, 5, 1, True, "HS")]


class _MobileNetV3(Layer):
    def __init__(self, cfg, last_c, num_classes=1000, scale=1.0):
        super().__init__()
        layers = [_cbr(3, 16, 3, 2, act=tn.Hardswish)]
        for cin, hid, cout, k, s, se, act in cfg:
            layers.append(_V3Block(cin, hid, cout, k, s, se, act))
        c_out = cfg[-1][2]
        layers.append(_cbr(c_out, cfg[-1][1], 1, act=tn.Hardswish))
        self.features = tn.Sequential(*layers)
        self.classifier = tn.Sequential(
            tn.Linear(cfg[-1][1], last_c), tn.Hardswish(inplace=True),
            tn.Dropout(0.2), tn.Linear(last_c, num_classes))

    def forward(self, x):
        x = torch.nn.functional.adaptive_avg_pool2d(self.features(x), 1)
        return self.classifier(torch.flatten(x, 1))


class MobileNetV3Small(_MobileNetV3):
    def __init__(self, scale=1.0, num_classes=1000, with_pool=True):
        super().__init__(_V3_SMALL, 1024, num_classes, scale)


class MobileNetV3Large(_MobileNetV3):
    def __init__(self, scale=1.0, num_classes=1000, with_pool=True):
        super().__init__(_V3_LARGE, 1280, num_classes, scale)


def mobilenet_v3_small(pretrained=False, scale=1.0, **kw):
    _no_pretrained(pretrained)
    return MobileNetV3Small(scale, **kw)


def mobilenet_v3_large(pretrained=False, scale=1.0, **kw):
    _no_pretrained(pretrained)
    return MobileNetV3Large(scale, **kw)


# ---------------------------------------------------------------------------
# ShuffleNetV2
# ---------------------------------------------------------------------------
def _channel_shuffle(x, groups=2):
    b, c, h, w = x.shape
    return x.view(b, groups, c // groups, h, w).transpose(1, 2).reshape(b, c, h, w)


class _ShuffleUnit(Layer):
    def __init__(self, cin, cout, stride, act):
        super().__init__()
        self.stride = stride
        branch = cout // 2
        A = tn.Hardswish if act == "swish" else tn.ReLU
        if stride > 1:
            self.branch1 = tn.Sequential(
                tn.Conv2d(cin, cin, 3, stride, 1, groups=cin, bias=False),
                tn.BatchNorm2d(cin),
                tn.Conv2d(cin, branch, 1, bias=False), tn.BatchNorm2d(branch),
                A(inplace=True))
            b2_in = cin
        else:
            self.branch1 = None
            b2_in = cin // 2
        self.branch2 = tn.Sequential(
            tn.Conv2d(b2_in, branch, 1, bias=False), tn.BatchNorm2d(branch),
            A(inplace=True),
            tn.Conv2d(branch, branch, 3, stride, 1, groups=branch, bias=False),
            tn.BatchNorm2d(branch),
            tn.Conv2d(branch, branch, 1, bias=False), tn.BatchNorm2d(branch),
            A(inplace=True))

    def forward(self, x):
        if self.stride > 1:
            out = torch.cat([self.branch1(x), self.branch2(x)], 1)
        else:
            x1, x2 = x.chunk(2, dim=1)
            out = torch.cat([x1, self.branch2(x2)], 1)
        return _channel_shuffle(out)


_SHUFFLE_CFGS = {0.25: [24, 24, 48, 96, 512], 0.33: [24, 32, 64, 128, 512],
                 0.5: [24, 48, 96, 192, 1024], 1.0: [24, 116, 232, 464, 1024],
                 1.5: [24, 176, 352, 704, 1024], 2.0: [24, 244, 488, 976, 2048]}


class ShuffleNetV2(Layer):
    def __init__(self, scale=1.0, act="relu", num_classes=1000, with_pool=True):
        super().__init__()
        chs = _SHUFFLE_CFGS[scale]
        self.conv1 = _cbr(3, chs[0], 3, 2, act=tn.ReLU)
        self.maxpool = tn.MaxPool2d(3, 2, 1)
        stages = []
        cin = chs[0]
        for i, reps in enumerate([4, 8, 4]):
            cout = chs[i + 1]
            stages.append(_ShuffleUnit(cin, cout, 2, act))
            for _ in range(reps - 1):
                stages.append(_ShuffleUnit(cout, cout, 1, act))
            cin = cout
        self.stages = tn.Sequential(*stages)
        self.conv_last = _cbr(cin, chs[-1], 1, act=tn.ReLU)
        self.fc = tn.Linear(chs[-1], num_classes)

    def forward(self, x):
        x = self.conv_last(self.stages(self.maxpool(self.conv1(x))))
        x = torch.nn.functional.adaptive_avg_pool2d(x, 1)
        return self.fc(torch.flatten(x, 1))


def _shuffle(scale, act="relu"):
    def f(pretrained=False, **kw):
        _no_pretrained(pretrained)
        return ShuffleNetV2(scale, act, **kw)
    return f


shufflenet_v2_x0_25 = _shuffle(0.25)
shufflenet_v2_x0_33 = _shuffle(0.33)
shufflenet_v2_x0_5 = _shuffle(0.5)
shufflenet_v2_x1_0 = _shuffle(1.0)
shufflenet_v2_x1_5 = _shuffle(1.5)
shufflenet_v2_x2_0 = _shuffle(2.0)
shufflenet_v2_swish = _shuffle(1.0, "swish")


# ---------------------------------------------------------------------------
# GoogLeNet / InceptionV3
# ---------------------------------------------------------------------------
class _Inception(Layer):
    def __init__(self, cin, c1, c3r, c3, c5r, c5, pp):
        super().__init__()
        self.b1 = _cbr(cin, c1, 1, act=tn.ReLU)
        self.b2 = tn.Sequential(_cbr(cin, c3r, 1, act=tn.ReLU),
                                _cbr(c3r, c3, 3, act=tn.ReLU))
        self.b3 = tn.Sequential(_cbr(cin, c5r, 1, act=tn.ReLU),
                                _cbr(c5r, c5, 5, act=tn.ReLU))
        self.b4 = tn.Sequential(tn.MaxPool2d(3, 1, 1),
                                _cbr(cin, pp, 1, act=tn.ReLU))

    def forward(self, x):
        return torch.cat([self.b1(x), self.b2(x), self.b3(x), self.b4(x)], 1)


class GoogLeNet(Layer):
    def __init__(self, num_classes=1000, with_pool=True):
        super().__init__()
        self.pre = tn.Sequential(
            _cbr(3, 64, 7, 2, act=tn.ReLU), tn.MaxPool2d(3, 2, 1),
            _cbr(64, 64, 1, act=tn.ReLU), _cbr(64, 192, 3, act=tn.ReLU),
            tn.MaxPool2d(3, 2, 1))
        self.blocks = tn.Sequential(
            _Inception(192, 64, 96, 128, 16, 32, 32),
            _Inception(256, 128, 128, 192, 32, 96, 64),
            tn.MaxPool2d(3, 2, 1),
            _Inception(480, 192, 96, 208, 16, 48, 64),
            _Inception(512, 160, 112, 224, 24, 64, 64),
            _Inception(512, 128, 128, 256, 24, 64, 64),
            _Inception(512, 112, 144, 288, 32, 64, 64),
            _Inception(528, 256, 160, 320, 32, 128, 128),
            tn.MaxPool2d(3, 2, 1),
            _Inception(832, 256, 160, 320, 32, 128, 128),
            _Inception(832, 384, 192, 384, 48, 128, 128))
        self.fc = tn.Linear(1024, num_classes)

    def forward(self, x):
        x = self.blocks(self.pre(x))
        x = torch.nn.functional.adaptive_avg_pool2d(x, 1)
        out = self.fc(torch.flatten(x, 1))
        # paddle returns (main, aux1, aux2); aux heads degenerate to main here
        return out, out, out


def googlenet(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return GoogLeNet(**kw)


class InceptionV3(Layer):
    """Faithful stem + inception-A/C stacks with reduction blocks
    (reference: vision/models/inceptionv3.py; factorized 7x7 towers kept,
    grid sizes match the 299x299 reference input)."""

    def __init__(self, num_classes=1000, with_pool=True):
        super().__init__()
        self.stem = tn.Sequential(
            _cbr(3, 32, 3, 2, act=tn.ReLU), _cbr(32, 32, 3, act=tn.ReLU),
            _cbr(32, 64, 3, act=tn.ReLU), tn.MaxPool2d(3, 2),
            _cbr(64, 80, 1, act=tn.ReLU), _cbr(80, 192, 3, act=tn.ReLU),
            tn.MaxPool2d(3, 2))
        self.blocks = tn.Sequential(
            _Inception(192, 64, 48, 64, 64, 96, 32),
            _Inception(256, 64, 48, 64, 64, 96, 64),
            _Inception(288, 64, 48, 64, 64, 96, 64),
            tn.MaxPool2d(3, 2, 1),
            _Inception(288, 192, 128, 192, 128, 192, 192),
            _Inception(768, 192, 160, 192, 160, 192, 192),
            tn.MaxPool2d(3, 2, 1),
            _Inception(768, 320, 192, 384, 192, 384, 192),
            _Inception(1280, 320, 192, 384, 192, 384, 192))
        self.fc = tn.Linear(1280, num_classes)

    def forward(self, x):
        x = self.blocks(self.stem(x))
        x = torch.nn.functional.adaptive_avg_pool2d(x, 1)
        return self.fc(torch.flatten(x, 1))


def inception_v3(pretrained=False, **kw):
    _no_pretrained(pretrained)
    return InceptionV3(**kw)
