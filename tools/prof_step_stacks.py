"""Attribute flagship-step copy/cast/add kernels to python lines
(torch.profiler with_stack) -- chases the ~8% glue in bench_6p7b stats."""
import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import paddle_amd as paddle
from paddle_amd.distributed.fleet.sharding import GroupShardedStage3, ShardedAdamW
from paddle_amd.models import GPTPretrainingCriterion, build_gpt

paddle.seed(0)
model = build_gpt("gpt3-6.7b", max_seq_len=2048).to(device="cuda", dtype=torch.bfloat16)
wrapped = GroupShardedStage3(model, device=torch.device("cuda"))
opt = ShardedAdamW(wrapped, learning_rate=1e-4, beta1=0.9, beta2=0.95,
                   epsilon=1e-8, weight_decay=0.1,
                   grad_clip=paddle.nn.ClipGradByGlobalNorm(1.0))
crit = GPTPretrainingCriterion()
ids = torch.randint(0, 50304, (4, 2048), device="cuda")
labels = torch.randint(0, 50304, (4, 2048), device="cuda")

def step():
    loss = crit(wrapped(ids), labels)
    loss.backward()
    opt.step()
    opt.clear_grad()

for _ in range(2):
    step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
import torch._C._profiler as _p
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             with_stack=True,
             experimental_config=_p._ExperimentalConfig(verbose=True)) as prof:
    step()
    torch.cuda.synchronize()
evs = prof.key_averages(group_by_stack_n=6)
rows = [e for e in evs
        if e.device_time_total > 0 and
        ("copy" in e.key.lower() or "Memcpy" in e.key or "to" == e.key[-2:].lower()
         or "_to_copy" in e.key or "cast" in e.key.lower()
         or "add" in e.key.lower() or "fill" in e.key.lower())]
rows.sort(key=lambda e: -e.device_time_total)
for e in rows[:12]:
    print(f"{e.device_time_total/1e3:8.2f}ms {e.count:5d}x  {e.key[:60]}")
    for ln in (e.stack or [])[:6]:
        if "paddle_amd" in ln or "bench" in ln:
            print("     ", ln[-110:])
