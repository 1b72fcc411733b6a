"""Secondary BASELINE configs measured on 1x MI355X (bf16, synthetic):

  - BERT-base pretraining step (MLM+NSP), fused attention/FFN HIP path
  - GPT-MoE (350M-scale dense trunk, 64 experts, top-2) training step

The north-star GPT-3-6.7B number comes from bench.py; these cover the
other rows of BASELINE.md's measurement table on a single GPU.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

import paddle_amd as paddle  # noqa: E402


def run(name, model, make_batch, loss_fn, steps=8, warmup=3, lr=1e-4):
    dev = torch.device("cuda")
    model = model.to(device=dev, dtype=torch.bfloat16)
    opt = paddle.optimizer.AdamW(learning_rate=lr, parameters=model.parameters(),
                                 grad_clip=paddle.nn.ClipGradByGlobalNorm(1.0))
    batch = make_batch(dev)

    def step():
        loss = loss_fn(model, batch)
        loss.backward()
        opt.step()
        opt.clear_grad()
        return loss

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        loss = step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    tokens = batch[0].numel()
    print(f"{name:34s} {dt * 1e3:8.2f} ms/step  {tokens / dt:12.0f} tokens/s"
          f"  loss={float(loss):.3f}  peak={torch.cuda.max_memory_allocated() / 2**30:.1f} GB")


def main():
    torch.manual_seed(0)

    # BERT-base: batch 64, seq 512 (classic pretraining shape)
    from paddle_amd.models import build_bert
    bert = build_bert("bert-base", max_position=512)
    B, S = 64, 512

    def bert_batch(dev):
        ids = torch.randint(0, 30522, (B, S), device=dev)
        mlm_labels = torch.randint(0, 30522, (B, S), device=dev)
        nsp = torch.randint(0, 2, (B,), device=dev)
        return ids, mlm_labels, nsp

    ce = torch.nn.functional.cross_entropy

    def bert_loss(m, batch):
        ids, mlm, nsp = batch
        mlm_logits, nsp_logits = m(ids)
        return (ce(mlm_logits.float().reshape(-1, 30522), mlm.reshape(-1)) +
                ce(nsp_logits.float(), nsp))

    run("bert-base b64 s512", bert, bert_batch, bert_loss)
    del bert
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()

    # GPT-MoE: 350M dense trunk + 64 experts top-2 (EP=1 on one GPU)
    from paddle_amd.models.gpt import GPTConfig
    from paddle_amd.models.moe import GPTMoEForPretraining
    cfg = GPTConfig(vocab_size=50304, hidden_size=1024, num_layers=12,
                    num_heads=16, intermediate_size=4096, max_seq_len=1024)
    moe = GPTMoEForPretraining(cfg, num_experts=64, k=2)
    Bm, Sm = 8, 1024

    def moe_batch(dev):
        ids = torch.randint(0, cfg.vocab_size, (Bm, Sm), device=dev)
        labels = torch.randint(0, cfg.vocab_size, (Bm, Sm), device=dev)
        return ids, labels

    from paddle_amd.models import GPTPretrainingCriterion
    crit = GPTPretrainingCriterion()

    def moe_loss(m, batch):
        ids, labels = batch
        return crit(m(ids), labels)

    run("gpt-moe-64e 350M-trunk b8 s1024", moe, moe_batch, moe_loss)
    # fp8 expert GEMMs (BASELINE config 5: expert-parallel + fp8 MFMA)
    from paddle_amd.incubate.fp8 import convert_experts_to_fp8
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    moe8 = GPTMoEForPretraining(cfg, num_experts=64, k=2)
    convert_experts_to_fp8(moe8)
    run("gpt-moe-64e fp8-experts", moe8, moe_batch, moe_loss)
    del moe8
    del moe
    torch.cuda.empty_cache()
    torch.cuda.reset_peak_memory_stats()
    bench_llama7b()


def bench_llama7b():
    """Llama-2-7B sharding-3 on one GPU (b4 s2048 bf16) -- the dense-model
    secondary config measured standalone (70B needs the 8-GPU hybrid run)."""
    import paddle_amd as paddle
    from paddle_amd.models import build_llama
    from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                       ShardedAdamW)
    dev = torch.device("cuda")
    model = build_llama("llama2-7b", max_seq_len=2048).to(dev, torch.bfloat16)
    wrapped = GroupShardedStage3(model, device=dev)
    opt = ShardedAdamW(wrapped, learning_rate=1e-4,
                       grad_clip=paddle.nn.ClipGradByGlobalNorm(1.0))
    B, S, V = 4, 2048, 32000
    ids = torch.randint(0, V, (B, S), device=dev)
    labels = torch.randint(0, V, (B, S), device=dev)

    import torch.nn.functional as F

    def step():
        logits = wrapped(ids)
        # causal shift: predict position t+1 (unshifted CE is trivially
        # memorizable from the visible input token)
        loss = F.cross_entropy(logits[:, :-1].float().reshape(-1, V),
                               labels[:, 1:].reshape(-1))
        loss.backward()
        opt.step()
        opt.clear_grad()
        return loss

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(6):
        loss = step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 6
    print(f"{'llama2-7b shard3 b4 s2048':34s} {dt * 1e3:8.2f} ms/step  "
          f"{B * S / dt:12.0f} tokens/s  loss={float(loss):.3f}  "
          f"peak={torch.cuda.max_memory_allocated() / 2**30:.1f} GB")


if __name__ == "__main__":
    main()
