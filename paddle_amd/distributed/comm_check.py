"""Collective-consistency (desync/race) checker.

Reference role: paddle's comm dynamic check
(paddle/phi/core/distributed/check/ -- per-collective shape/dtype cross-
rank validation) and the comm-task watchdog's desync report.

When enabled, every collective issued through paddle_amd.distributed is
recorded as (op, shape, dtype) in a per-rank journal; `verify()` (or the
automatic every-N check) all_gathers the journal hashes and raises on
divergence -- catching the classic hangs-forever bugs (ranks calling
different collectives, different orders, or mismatched shapes) while
still on a live process group.
"""
from __future__ import annotations

import hashlib

_enabled = False
_journal: list[str] = []
_auto_every = 0


def enable(auto_check_every=0):
    """Start recording collectives; auto-verify every N records if set."""
    global _enabled, _auto_every
    _enabled = True
    _auto_every = auto_check_every


def disable():
    global _enabled
    _enabled = False
    _journal.clear()


def record(op, tensor=None, extra=""):
    if not _enabled:
        return
    desc = op
    if tensor is not None and hasattr(tensor, "shape"):
        desc += f":{tuple(tensor.shape)}:{tensor.dtype}"
    if extra:
        desc += f":{extra}"
    _journal.append(desc)
    if _auto_every and len(_journal) % _auto_every == 0:
        verify()


def journal():
    return list(_journal)


def _digest():
    h = hashlib.sha1()
    for d in _journal:
        h.update(d.encode())
    return h.hexdigest(), len(_journal)


def verify(group=None):
    """Cross-rank check: every rank must have issued the same collective
    sequence.  Raises RuntimeError naming the first divergent entry."""
    from . import collective as C
    if not C.is_initialized():
        return True
    import torch.distributed as dist
    world = dist.get_world_size()
    if world == 1:
        return True
    dig, n = _digest()
    gathered = [None] * world
    dist.all_gather_object(gathered, (dig, n))
    if all(g == gathered[0] for g in gathered):
        return True
    # divergence: exchange full journals to locate it
    full = [None] * world
    dist.all_gather_object(full, _journal)
    limit = min(len(j) for j in full)
    for i in range(limit):
        entries = {j[i] for j in full}
        if len(entries) > 1:
            raise RuntimeError(
                f"collective desync at call #{i}: ranks disagree: "
                + "; ".join(f"rank{r}={full[r][i]}" for r in range(world)))
    counts = [len(j) for j in full]
    raise RuntimeError(
        f"collective desync: ranks issued different call counts {counts}; "
        f"first {limit} calls agree")
