"""paddle.version parity (reference: generated python/paddle/version/
__init__.py -- full_version/major/minor/patch/commit, cuda()/cudnn()/
show()).  Reports the ROCm/HIP stack this build targets."""
from __future__ import annotations

import torch

full_version = "0.1.0"
major, minor, patch = "0", "1", "0"
rc = "0"
commit = "mi355x-native"
istaged = True
with_pip_cuda_libraries = "OFF"


def show():
    print(f"paddle_amd {full_version} (commit {commit})")
    print(f"hip: {torch.version.hip}")
    print("arch: gfx950 (MI355X/CDNA4)")


def cuda():
    """Reference API name; returns the HIP runtime version on ROCm."""
    return torch.version.hip or "False"


def cudnn():
    """MIOpen stands in for cudnn on this stack."""
    try:
        return str(torch.backends.cudnn.version())
    except Exception:
        return "False"


def nccl():
    try:
        return ".".join(str(x) for x in torch.cuda.nccl.version())
    except Exception:
        return "rccl"


def xpu():
    return "False"


def xpu_xccl():
    return "False"


def cinn():
    return "False"
