"""Print hipBLASLt heuristic support for each epilogue at bench shapes."""
import torch
import paddle_amd._C as C

names = {4: "BIAS", 32: "GELU", 36: "GELU_BIAS", 160: "GELU_AUX",
         164: "GELU_AUX_BIAS", 192: "DGELU", 208: "DGELU_BGRAD",
         256: "BGRADA", 512: "BGRADB"}
for (m, n, k) in [(1024, 1024, 1024), (16384, 16384, 4096)]:
    row = []
    for epi, nm in names.items():
        try:
            cnt = C.lt_epilogue_probe(m, n, k, epi)
        except RuntimeError as e:
            cnt = f"ERR"
        row.append(f"{nm}={cnt}")
    print(f"{m}x{n}x{k}: " + " ".join(row))
