"""paddle.vision.transforms (reference: python/paddle/vision/transforms/
{transforms,functional}.py).

Tensor-native implementation: every transform works on CHW float tensors
(or HWC uint8 numpy arrays, converted on entry) -- no PIL dependency, so
the pipeline runs on device memory and never round-trips through Python
imaging objects.
"""
from __future__ import annotations

import math
import random as _random

import numpy as np
import torch
import torch.nn.functional as F


# ---------------------------------------------------------------------------
# functional API
# ---------------------------------------------------------------------------
def _as_chw(img):
    """numpy HWC (uint8 or float) -> float CHW tensor; tensors pass through."""
    if isinstance(img, np.ndarray):
        t = torch.from_numpy(img)
        if t.dim() == 2:
            t = t.unsqueeze(-1)
        t = t.permute(2, 0, 1)
        if t.dtype == torch.uint8:
            t = t.float() / 255.0
        return t.float()
    return img


def to_tensor(pic, data_format="CHW"):
    t = _as_chw(pic)
    if data_format == "HWC":
        t = t.permute(1, 2, 0)
    return t


def resize(img, size, interpolation="bilinear"):
    img = _as_chw(img)
    if isinstance(size, int):
        c, h, w = img.shape[-3:]
        if h <= w:
            size = (size, int(size * w / h))
        else:
            size = (int(size * h / w), size)
    mode = {"nearest": "nearest", "bilinear": "bilinear",
            "bicubic": "bicubic"}.get(interpolation, "bilinear")
    ac = None if mode == "nearest" else False
    return F.interpolate(img.unsqueeze(0), size=tuple(size), mode=mode,
                         align_corners=ac).squeeze(0)


def crop(img, top, left, height, width):
    return _as_chw(img)[..., top:top + height, left:left + width]


def center_crop(img, output_size):
    img = _as_chw(img)
    if isinstance(output_size, int):
        output_size = (output_size, output_size)
    h, w = img.shape[-2:]
    th, tw = output_size
    return crop(img, (h - th) // 2, (w - tw) // 2, th, tw)


def hflip(img):
    return _as_chw(img).flip(-1)


def vflip(img):
    return _as_chw(img).flip(-2)


def pad(img, padding, fill=0, padding_mode="constant"):
    img = _as_chw(img)
    if isinstance(padding, int):
        padding = [padding] * 4
    elif len(padding) == 2:
        padding = [padding[0], padding[0], padding[1], padding[1]]
    mode = {"constant": "constant", "edge": "replicate",
            "reflect": "reflect", "symmetric": "reflect"}[padding_mode]
    kw = {"value": fill} if mode == "constant" else {}
    return F.pad(img.unsqueeze(0), list(padding), mode=mode, **kw).squeeze(0)


def normalize(img, mean, std, data_format="CHW", to_rgb=False):
    img = _as_chw(img) if data_format == "CHW" else torch.as_tensor(img).float()
    mean = torch.as_tensor(mean, dtype=img.dtype, device=img.device)
    std = torch.as_tensor(std, dtype=img.dtype, device=img.device)
    if data_format == "CHW":
        return (img - mean.view(-1, 1, 1)) / std.view(-1, 1, 1)
    return (img - mean) / std


def adjust_brightness(img, brightness_factor):
    return (_as_chw(img) * brightness_factor).clamp(0, 1)


def adjust_contrast(img, contrast_factor):
    img = _as_chw(img)
    mean = img.mean((-3, -2, -1), keepdim=True)
    return ((img - mean) * contrast_factor + mean).clamp(0, 1)


def adjust_saturation(img, saturation_factor):
    img = _as_chw(img)
    gray = to_grayscale(img).expand_as(img)
    return ((img - gray) * saturation_factor + gray).clamp(0, 1)


def adjust_hue(img, hue_factor):
    """Shift hue by hue_factor (in [-0.5, 0.5]) via RGB<->HSV."""
    img = _as_chw(img)
    r, g, b = img[0], img[1], img[2]
    mx, _ = img.max(0)
    mn, _ = img.min(0)
    d = (mx - mn).clamp(min=1e-8)
    h = torch.zeros_like(mx)
    m = mx == r
    h[m] = ((g - b) / d)[m] % 6
    m = mx == g
    h[m] = ((b - r) / d + 2)[m]
    m = mx == b
    h[m] = ((r - g) / d + 4)[m]
    h = (h / 6 + hue_factor) % 1.0
    s = torch.where(mx > 0, d / mx.clamp(min=1e-8), torch.zeros_like(mx))
    v = mx
    # HSV -> RGB
    h6 = h * 6
    i = h6.floor()
    f = h6 - i
    p = v * (1 - s)
    q = v * (1 - f * s)
    t = v * (1 - (1 - f) * s)
    i = i.long() % 6
    out = torch.stack([
        torch.where(i == 0, v, torch.where(i == 1, q, torch.where(
            i == 2, p, torch.where(i == 3, p, torch.where(i == 4, t, v))))),
        torch.where(i == 0, t, torch.where(i == 1, v, torch.where(
            i == 2, v, torch.where(i == 3, q, torch.where(i == 4, p, p))))),
        torch.where(i == 0, p, torch.where(i == 1, p, torch.where(
            i == 2, t, torch.where(i == 3, v, torch.where(i == 4, v, q))))),
    ])
    return out


def to_grayscale(img, num_output_channels=1):
    img = _as_chw(img)
    w = torch.tensor([0.299, 0.587, 0.114], device=img.device)
    gray = (img[:3] * w.view(3, 1, 1)).sum(0, keepdim=True)
    return gray.expand(num_output_channels, *gray.shape[1:]) \
        if num_output_channels > 1 else gray


def _affine_mat(angle, translate, scale, shear, h, w):
    rot = math.radians(angle)
    sx, sy = (math.radians(s) for s in (shear if isinstance(shear, (list, tuple))
                                        else (shear, 0.0)))
    a = math.cos(rot - sy) / math.cos(sy)
    b = -math.cos(rot - sy) * math.tan(sx) / math.cos(sy) - math.sin(rot)
    c = math.sin(rot - sy) / math.cos(sy)
    d = -math.sin(rot - sy) * math.tan(sx) / math.cos(sy) + math.cos(rot)
    m = torch.tensor([[a, b, 0.0], [c, d, 0.0]]) / scale
    m[0, 2] = -2.0 * translate[0] / w
    m[1, 2] = -2.0 * translate[1] / h
    return m


def affine(img, angle, translate, scale, shear, interpolation="bilinear",
           fill=0, center=None):
    img = _as_chw(img)
    h, w = img.shape[-2:]
    theta = _affine_mat(angle, translate, scale, shear, h, w).unsqueeze(0)
    grid = F.affine_grid(theta, [1, img.shape[0], h, w], align_corners=False)
    return F.grid_sample(img.unsqueeze(0), grid, mode=interpolation,
                         align_corners=False).squeeze(0)


def rotate(img, angle, interpolation="nearest", expand=False, center=None,
           fill=0):
    return affine(img, angle, (0, 0), 1.0, (0.0, 0.0),
                  interpolation="bilinear" if interpolation != "nearest"
                  else "nearest")


def perspective(img, startpoints, endpoints, interpolation="nearest", fill=0):
    """4-point perspective warp via the homography solved in least squares."""
    img = _as_chw(img)
    h, w = img.shape[-2:]
    a = []
    bvec = []
    for (sx, sy), (ex, ey) in zip(startpoints, endpoints):
        a.append([ex, ey, 1, 0, 0, 0, -sx * ex, -sx * ey])
        a.append([0, 0, 0, ex, ey, 1, -sy * ex, -sy * ey])
        bvec += [sx, sy]
    A = torch.tensor(a, dtype=torch.float64)
    B = torch.tensor(bvec, dtype=torch.float64)
    hvec = torch.linalg.lstsq(A, B).solution
    H = torch.cat([hvec, torch.ones(1, dtype=torch.float64)]).reshape(3, 3).float()
    ys, xs = torch.meshgrid(torch.arange(h), torch.arange(w), indexing="ij")
    ones = torch.ones_like(xs)
    pts = torch.stack([xs, ys, ones], 0).reshape(3, -1).float()
    src = H @ pts
    src = src[:2] / src[2:].clamp(min=1e-8)
    gx = src[0].reshape(h, w) / (w - 1) * 2 - 1
    gy = src[1].reshape(h, w) / (h - 1) * 2 - 1
    grid = torch.stack([gx, gy], -1).unsqueeze(0)
    mode = "bilinear" if interpolation != "nearest" else "nearest"
    return F.grid_sample(img.unsqueeze(0), grid, mode=mode,
                         align_corners=True).squeeze(0)


def erase(img, i, j, h, w, v, inplace=False):
    img = _as_chw(img)
    if not inplace:
        img = img.clone()
    img[..., i:i + h, j:j + w] = v
    return img


# ---------------------------------------------------------------------------
# transform classes
# ---------------------------------------------------------------------------
class BaseTransform:
    def __init__(self, keys=None):
        self.keys = keys

    def __call__(self, img):
        return self._apply_image(img)

    def _apply_image(self, img):  # pragma: no cover - abstract
        raise NotImplementedError


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class ToTensor(BaseTransform):
    def __init__(self, data_format="CHW", keys=None):
        super().__init__(keys)
        self.data_format = data_format

    def _apply_image(self, img):
        return to_tensor(img, self.data_format)


class Normalize(BaseTransform):
    def __init__(self, mean=0.0, std=1.0, data_format="CHW", to_rgb=False,
                 keys=None):
        super().__init__(keys)
        self.mean = [mean] * 3 if isinstance(mean, (int, float)) else mean
        self.std = [std] * 3 if isinstance(std, (int, float)) else std
        self.data_format = data_format

    def _apply_image(self, img):
        return normalize(img, self.mean, self.std, self.data_format)


class Resize(BaseTransform):
    def __init__(self, size, interpolation="bilinear", keys=None):
        super().__init__(keys)
        self.size = size
        self.interpolation = interpolation

    def _apply_image(self, img):
        return resize(img, self.size, self.interpolation)


class CenterCrop(BaseTransform):
    def __init__(self, size, keys=None):
        super().__init__(keys)
        self.size = size

    def _apply_image(self, img):
        return center_crop(img, self.size)


class RandomCrop(BaseTransform):
    def __init__(self, size, padding=None, pad_if_needed=False, fill=0,
                 padding_mode="constant", keys=None):
        super().__init__(keys)
        self.size = (size, size) if isinstance(size, int) else size
        self.padding = padding

    def _apply_image(self, img):
        img = _as_chw(img)
        if self.padding:
            img = pad(img, self.padding)
        h, w = img.shape[-2:]
        th, tw = self.size
        top = _random.randint(0, max(0, h - th))
        left = _random.randint(0, max(0, w - tw))
        return crop(img, top, left, th, tw)


class RandomResizedCrop(BaseTransform):
    def __init__(self, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3),
                 interpolation="bilinear", keys=None):
        super().__init__(keys)
        self.size = (size, size) if isinstance(size, int) else size
        self.scale = scale
        self.ratio = ratio
        self.interpolation = interpolation

    def _apply_image(self, img):
        img = _as_chw(img)
        h, w = img.shape[-2:]
        area = h * w
        for _ in range(10):
            ta = area * _random.uniform(*self.scale)
            ar = math.exp(_random.uniform(math.log(self.ratio[0]),
                                          math.log(self.ratio[1])))
            cw = int(round(math.sqrt(ta * ar)))
            ch = int(round(math.sqrt(ta / ar)))
            if 0 < cw <= w and 0 < ch <= h:
                top = _random.randint(0, h - ch)
                left = _random.randint(0, w - cw)
                return resize(crop(img, top, left, ch, cw), self.size,
                              self.interpolation)
        return resize(center_crop(img, min(h, w)), self.size, self.interpolation)


class RandomHorizontalFlip(BaseTransform):
    def __init__(self, prob=0.5, keys=None):
        super().__init__(keys)
        self.prob = prob

    def _apply_image(self, img):
        return hflip(img) if _random.random() < self.prob else _as_chw(img)


class RandomVerticalFlip(BaseTransform):
    def __init__(self, prob=0.5, keys=None):
        super().__init__(keys)
        self.prob = prob

    def _apply_image(self, img):
        return vflip(img) if _random.random() < self.prob else _as_chw(img)


class Transpose(BaseTransform):
    def __init__(self, order=(2, 0, 1), keys=None):
        super().__init__(keys)
        self.order = order

    def _apply_image(self, img):
        if isinstance(img, np.ndarray):
            return img.transpose(self.order)
        return img.permute(self.order)


class BrightnessTransform(BaseTransform):
    def __init__(self, value, keys=None):
        super().__init__(keys)
        self.value = value

    def _apply_image(self, img):
        f = _random.uniform(max(0, 1 - self.value), 1 + self.value)
        return adjust_brightness(img, f)


class ContrastTransform(BrightnessTransform):
    def _apply_image(self, img):
        f = _random.uniform(max(0, 1 - self.value), 1 + self.value)
        return adjust_contrast(img, f)


class SaturationTransform(BrightnessTransform):
    def _apply_image(self, img):
        f = _random.uniform(max(0, 1 - self.value), 1 + self.value)
        return adjust_saturation(img, f)


class HueTransform(BrightnessTransform):
    def _apply_image(self, img):
        f = _random.uniform(-self.value, self.value)
        return adjust_hue(img, f)


class ColorJitter(BaseTransform):
    def __init__(self, brightness=0, contrast=0, saturation=0, hue=0, keys=None):
        super().__init__(keys)
        self.ts = []
        if brightness:
            self.ts.append(BrightnessTransform(brightness))
        if contrast:
            self.ts.append(ContrastTransform(contrast))
        if saturation:
            self.ts.append(SaturationTransform(saturation))
        if hue:
            self.ts.append(HueTransform(hue))

    def _apply_image(self, img):
        order = list(self.ts)
        _random.shuffle(order)
        for t in order:
            img = t(img)
        return img


class Pad(BaseTransform):
    def __init__(self, padding, fill=0, padding_mode="constant", keys=None):
        super().__init__(keys)
        self.padding, self.fill, self.mode = padding, fill, padding_mode

    def _apply_image(self, img):
        return pad(img, self.padding, self.fill, self.mode)


class RandomAffine(BaseTransform):
    def __init__(self, degrees, translate=None, scale=None, shear=None,
                 interpolation="nearest", fill=0, center=None, keys=None):
        super().__init__(keys)
        self.degrees = (-degrees, degrees) if isinstance(degrees, (int, float)) \
            else degrees
        self.translate = translate
        self.scale = scale
        self.shear = shear

    def _apply_image(self, img):
        img = _as_chw(img)
        h, w = img.shape[-2:]
        ang = _random.uniform(*self.degrees)
        tr = (0, 0)
        if self.translate:
            tr = (_random.uniform(-self.translate[0], self.translate[0]) * w,
                  _random.uniform(-self.translate[1], self.translate[1]) * h)
        sc = _random.uniform(*self.scale) if self.scale else 1.0
        sh = _random.uniform(*self.shear) if self.shear else 0.0
        return affine(img, ang, tr, sc, (sh, 0.0))


class RandomRotation(BaseTransform):
    def __init__(self, degrees, interpolation="nearest", expand=False,
                 center=None, fill=0, keys=None):
        super().__init__(keys)
        self.degrees = (-degrees, degrees) if isinstance(degrees, (int, float)) \
            else degrees

    def _apply_image(self, img):
        return rotate(img, _random.uniform(*self.degrees))


class RandomPerspective(BaseTransform):
    def __init__(self, prob=0.5, distortion_scale=0.5, interpolation="nearest",
                 fill=0, keys=None):
        super().__init__(keys)
        self.prob = prob
        self.distortion_scale = distortion_scale

    def _apply_image(self, img):
        img = _as_chw(img)
        if _random.random() >= self.prob:
            return img
        h, w = img.shape[-2:]
        d = self.distortion_scale
        def j(mx):
            return _random.randint(0, int(mx * d))
        start = [(0, 0), (w - 1, 0), (w - 1, h - 1), (0, h - 1)]
        end = [(j(w // 2), j(h // 2)),
               (w - 1 - j(w // 2), j(h // 2)),
               (w - 1 - j(w // 2), h - 1 - j(h // 2)),
               (j(w // 2), h - 1 - j(h // 2))]
        return perspective(img, start, end)


class Grayscale(BaseTransform):
    def __init__(self, num_output_channels=1, keys=None):
        super().__init__(keys)
        self.n = num_output_channels

    def _apply_image(self, img):
        return to_grayscale(img, self.n)


class RandomErasing(BaseTransform):
    def __init__(self, prob=0.5, scale=(0.02, 0.33), ratio=(0.3, 3.3),
                 value=0, inplace=False, keys=None):
        super().__init__(keys)
        self.prob, self.scale, self.ratio, self.value = prob, scale, ratio, value

    def _apply_image(self, img):
        img = _as_chw(img)
        if _random.random() >= self.prob:
            return img
        h, w = img.shape[-2:]
        area = h * w
        for _ in range(10):
            ta = area * _random.uniform(*self.scale)
            ar = _random.uniform(*self.ratio)
            eh, ew = int(round(math.sqrt(ta * ar))), int(round(math.sqrt(ta / ar)))
            if eh < h and ew < w:
                top = _random.randint(0, h - eh)
                left = _random.randint(0, w - ew)
                return erase(img, top, left, eh, ew, self.value)
        return img
