"""paddle.cost_model (reference: python/paddle/cost_model/cost_model.py).
Static-profiling cost model; here backed by eager timing of the callable
(the PIR program profiler is replaced by direct measurement)."""
from __future__ import annotations

import time


class CostModel:
    def profile_measure(self, program=None, startup=None, device="gpu",
                        fetch_cost_list=("time",), fn=None, iters=10):
        """Measure wall time of a callable (or a static Program's
        train step) and return {op: cost} style dict."""
        import torch
        target = fn
        if target is None and program is not None:
            from . import static as S
            exe = S.Executor()

            def target():
                exe.run(program)
        if target is None:
            return {}
        for _ in range(2):
            target()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            target()
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        return {"time": (time.perf_counter() - t0) / iters * 1000.0}
