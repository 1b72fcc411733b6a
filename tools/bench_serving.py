"""Serving throughput: continuous batching on GPT-350M-ish config, 1 GPU.

Measures decode tokens/s and mean TTFT with staggered arrivals through
the paged-KV engine.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import paddle_amd as paddle  # noqa: E402
from paddle_amd.models import build_gpt  # noqa: E402
from paddle_amd.models.gpt import GPTConfig, GPTForPretraining  # noqa: E402
from paddle_amd.serving import Engine, GPTModelRunner, Request  # noqa: E402


def main():
    paddle.seed(0)
    cfg = GPTConfig(vocab_size=50304, hidden_size=1024, num_layers=24,
                    num_heads=16, intermediate_size=4096, max_seq_len=2048)
    m = GPTForPretraining(cfg).to("cuda", torch.bfloat16)
    runner = GPTModelRunner(m, num_blocks=4096, block_size=16)
    eng = Engine(runner, num_blocks=4096, block_size=16, max_batch=32)
    import random
    random.seed(0)
    n_req, prompt_len, gen_len = 64, 128, 128
    for _ in range(n_req):
        eng.add_request(Request(
            prompt_ids=[random.randrange(cfg.vocab_size) for _ in range(prompt_len)],
            max_new_tokens=gen_len))
    t0 = time.perf_counter()
    eng.run_until_done()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    s = eng.stats()
    print(f"serving gpt-350M: {n_req} reqs x (p{prompt_len}+g{gen_len}) in {dt:.1f}s"
          f"  decode {s['output_tokens'] / dt:.0f} tok/s"
          f"  ttft {s['mean_ttft_s'] * 1e3:.0f} ms"
          f"  peak {torch.cuda.max_memory_allocated() / 2**30:.1f} GB")


if __name__ == "__main__":
    main()
