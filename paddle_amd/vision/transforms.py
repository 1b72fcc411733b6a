"""paddle.vision.transforms parity subset."""
from __future__ import annotations

import numpy as np
import torch


class Compose:
    def __init__(self, transforms):
        self.transforms = transforms

    def __call__(self, x):
        for t in self.transforms:
            x = t(x)
        return x


class ToTensor:
    def __init__(self, data_format="CHW"):
        self.data_format = data_format

    def __call__(self, img):
        a = np.asarray(img, dtype=np.float32) / 255.0
        if a.ndim == 2:
            a = a[None]
        elif self.data_format == "CHW" and a.shape[-1] in (1, 3, 4):
            a = a.transpose(2, 0, 1)
        return torch.from_numpy(a.copy())


class Normalize:
    def __init__(self, mean, std, data_format="CHW", to_rgb=False):
        self.mean = np.asarray(mean, dtype=np.float32)
        self.std = np.asarray(std, dtype=np.float32)

    def __call__(self, x):
        if isinstance(x, torch.Tensor):
            m = torch.as_tensor(self.mean).view(-1, 1, 1)
            s = torch.as_tensor(self.std).view(-1, 1, 1)
            return (x - m) / s
        return (np.asarray(x, dtype=np.float32) - self.mean) / self.std
