"""paddle.summary / paddle.flops (reference: python/paddle/hapi/
model_summary.py:36 summary, dynamic_flops.py flops)."""
from __future__ import annotations

import torch


def summary(net, input_size=None, dtypes=None, input=None):
    rows = []
    total = trainable = 0
    for name, p in net.named_parameters():
        n = p.numel()
        total += n
        if p.requires_grad:
            trainable += n
        rows.append((name, list(p.shape), n))
    width = max((len(r[0]) for r in rows), default=20) + 2
    lines = [f"{'Layer (param)':<{width}}{'Shape':<20}{'Param #':>12}",
             "-" * (width + 32)]
    for name, shape, n in rows:
        lines.append(f"{name:<{width}}{str(shape):<20}{n:>12,}")
    lines += ["-" * (width + 32),
              f"Total params: {total:,}",
              f"Trainable params: {trainable:,}",
              f"Non-trainable params: {total - trainable:,}"]
    print("\n".join(lines))
    return {"total_params": total, "trainable_params": trainable}


def flops(net, input_size, custom_ops=None, print_detail=False):
    """Linear/conv MAC counting via forward hooks on a dummy pass."""
    counts = [0]
    hooks = []

    def linear_hook(mod, inp, out):
        counts[0] += 2 * inp[0].numel() // inp[0].shape[-1] * \
            mod.weight.numel() // (1 if mod.weight.dim() < 2 else 1)

    def conv_hook(mod, inp, out):
        counts[0] += 2 * out.numel() * mod.weight.numel() // mod.weight.shape[0]

    for m in net.modules() if hasattr(net, "modules") else []:
        if isinstance(m, torch.nn.Linear):
            hooks.append(m.register_forward_hook(linear_hook))
        elif isinstance(m, (torch.nn.Conv1d, torch.nn.Conv2d)):
            hooks.append(m.register_forward_hook(conv_hook))
        elif type(m).__name__ == "Linear" and hasattr(m, "weight"):
            hooks.append(m.register_forward_hook(
                lambda mod, inp, out: counts.__setitem__(
                    0, counts[0] + 2 * (inp[0].numel() // inp[0].shape[-1]) *
                    mod.weight.numel())))
    try:
        x = torch.zeros(input_size)
        net(x)
    finally:
        for h in hooks:
            h.remove()
    if print_detail:
        print(f"FLOPs: {counts[0]:,}")
    return counts[0]
