"""Verify mfma_scale_f32_16x16x128_f8f6f4 fragment layout + scale encoding."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from paddle_amd import _ext
C = _ext.get_ext()
torch.manual_seed(0)
a = torch.randn(16, 128) * 0.5
b = torch.randn(128, 16) * 0.5
a8 = a.to(torch.float8_e4m3fn)
bt8 = b.t().contiguous().to(torch.float8_e4m3fn)
ref = a8.float() @ bt8.float().t()
one = 0x7F7F7F7F
out = C.mfma_probe_fp8mx(a8.view(torch.uint8).cuda(), bt8.view(torch.uint8).cuda(), one, one).cpu()
err = (out - ref).abs().max().item() / ref.abs().max().item()
print("layout relerr (scale=1):", err)
# scale test: sa = 2.0 blocks (e8m0 128 = 2^1)
two = 0x80808080
out2 = C.mfma_probe_fp8mx(a8.view(torch.uint8).cuda(), bt8.view(torch.uint8).cuda(), two, one).cpu()
ratio = (out2 / out).nanmedian().item()
print("A-scale=2 ratio:", ratio)
out4 = C.mfma_probe_fp8mx(a8.view(torch.uint8).cuda(), bt8.view(torch.uint8).cuda(), two, two).cpu()
print("both-scale=2 ratio:", (out4 / out).nanmedian().item())
