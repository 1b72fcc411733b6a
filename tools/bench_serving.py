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


def run_one(tag, model, n_req=64, prompt_len=128, gen_len=128, max_batch=32,
            num_blocks=4096, weight_only=False, runner_cls="gpt"):
    cls = GPTModelRunner
    if runner_cls == "llama":
        from paddle_amd.serving import LlamaModelRunner
        cls = LlamaModelRunner
    runner = cls(model, num_blocks=num_blocks, block_size=16,
                 weight_only=weight_only)
    runner.precapture((max_batch,))    # decode graph capture out of ttft
    eng = Engine(runner, num_blocks=num_blocks, block_size=16,
                 max_batch=max_batch)
    import random
    random.seed(0)
    V = model.cfg.vocab_size
    for _ in range(n_req):
        eng.add_request(Request(
            prompt_ids=[random.randrange(V) for _ in range(prompt_len)],
            max_new_tokens=gen_len))
    t0 = time.perf_counter()
    eng.step()                      # prefill wave (all same-length prompts)
    torch.cuda.synchronize()
    t1 = time.perf_counter()
    tok0 = sum(len(r.out_ids) for r in (eng.active + eng.completed + eng.waiting))
    n_steps = 0
    while eng.active or eng.waiting:
        eng.step()
        n_steps += 1
        if n_steps > 200000:
            break
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    dt_dec = time.perf_counter() - t1
    s = eng.stats()
    dec_toks = s["output_tokens"] - tok0
    print(f"serving {tag}: {n_req} reqs x (p{prompt_len}+g{gen_len}) in {dt:.1f}s"
          f"  decode {s['output_tokens'] / dt:.0f} tok/s (incl prefill)"
          f"  steady-decode {dec_toks / dt_dec:.0f} tok/s"
          f" ({dt_dec / max(n_steps,1) * 1e3:.2f} ms/step)"
          f"  ttft {s['mean_ttft_s'] * 1e3:.0f} ms"
          f"  peak {torch.cuda.max_memory_allocated() / 2**30:.1f} GB")


def main():
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt-350m",
                    choices=["gpt-350m", "gpt3-6.7b", "llama2-7b"])
    ap.add_argument("--weight-only", action="store_true",
                    help="int8 weight-only decode (MFMA W-streamer)")
    args = ap.parse_args()
    paddle.seed(0)
    tag_sfx = " int8-wo" if args.weight_only else ""
    if args.model == "gpt-350m":
        cfg = GPTConfig(vocab_size=50304, hidden_size=1024, num_layers=24,
                        num_heads=16, intermediate_size=4096, max_seq_len=2048)
        m = GPTForPretraining(cfg).to("cuda", torch.bfloat16)
        run_one("gpt-350M" + tag_sfx, m, weight_only=args.weight_only)
    elif args.model == "gpt3-6.7b":
        m = build_gpt("gpt3-6.7b", max_seq_len=2048).to("cuda", torch.bfloat16)
        run_one("gpt3-6.7B" + tag_sfx, m, n_req=64, max_batch=32,
                num_blocks=8192, weight_only=args.weight_only)
    else:
        from paddle_amd.models.llama import build_llama
        m = build_llama("llama2-7b", max_seq_len=2048).to("cuda", torch.bfloat16)
        run_one("llama2-7B" + tag_sfx, m, n_req=64, max_batch=32,
                num_blocks=8192, weight_only=args.weight_only,
                runner_cls="llama")


if __name__ == "__main__":
    main()
