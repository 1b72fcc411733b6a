"""Multi-process CPU (gloo, world_size=2) distributed tests -- the comm
semantics layer that must be correct by construction before GPU runs
(reference test style: test/collective/ subprocess harnesses)."""
import pytest

from dist_util import run_dist


def test_allreduce_allgather_broadcast():
    run_dist("""
        import torch
        import paddle_amd as paddle
        env = paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        w = paddle.distributed.get_world_size()
        assert w == 2
        t = torch.full((4,), float(r + 1))
        paddle.distributed.all_reduce(t)
        assert t.tolist() == [3.0] * 4, t
        outs = []
        paddle.distributed.all_gather(outs, torch.full((2,), float(r)))
        assert outs[0].tolist() == [0.0, 0.0] and outs[1].tolist() == [1.0, 1.0]
        b = torch.full((3,), float(r))
        paddle.distributed.broadcast(b, src=0)
        assert b.tolist() == [0.0] * 3
        # reduce_scatter (gloo emulation path)
        out = torch.zeros(2)
        paddle.distributed.reduce_scatter_tensor(out, torch.arange(4.0) + r)
        expect = [0 + 1 + 0 + 1, 1 + 2 + 1 + 2][0] if False else None
        # sum of both ranks: [1, 3, 5, 7]; rank slice:
        exp = [1.0, 3.0] if r == 0 else [5.0, 7.0]
        assert out.tolist() == exp, (out, exp)
        paddle.distributed.barrier()
    """)


def test_send_recv_and_objects():
    run_dist("""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        if r == 0:
            paddle.distributed.send(torch.arange(3.0), dst=1)
        else:
            t = torch.zeros(3)
            paddle.distributed.recv(t, src=0)
            assert t.tolist() == [0.0, 1.0, 2.0]
        objs = []
        paddle.distributed.all_gather_object(objs, {"rank": r})
        assert objs[0]["rank"] == 0 and objs[1]["rank"] == 1
    """)


def test_new_group_and_alltoall():
    run_dist("""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        g = paddle.distributed.new_group([0, 1])
        t = torch.full((2,), float(r))
        paddle.distributed.all_reduce(t, group=g)
        assert t.tolist() == [1.0, 1.0]
        ins = [torch.full((2,), float(r * 2 + i)) for i in range(2)]
        outs = []
        paddle.distributed.alltoall(ins, outs)
        # rank0 gets [r0 chunk0, r1 chunk0] = [0, 2]; rank1 [1, 3]
        exp = [0.0, 2.0] if r == 0 else [1.0, 3.0]
        assert [outs[0][0].item(), outs[1][0].item()] == exp
    """)


def test_data_parallel_grad_sync():
    run_dist("""
        import torch
        import paddle_amd as paddle
        from paddle_amd import nn
        paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        paddle.seed(100 + r)  # different init; DP broadcast must sync
        m = nn.Linear(4, 4)
        dp = paddle.distributed.DataParallel(m)
        # params synced from rank 0
        sd = {k: v.clone() for k, v in m.state_dict().items()}
        paddle.seed(7)  # same data
        x = paddle.randn([4, 4])
        # rank-dependent target -> different grads pre-sync
        y = x * (r + 1)
        loss = (dp(x) - y).square().mean()
        loss.backward()
        # after sync, grads identical on both ranks
        g = m.weight.grad.clone()
        gs = []
        paddle.distributed.all_gather(gs, g)
        assert torch.allclose(gs[0], gs[1], atol=1e-6), (gs[0] - gs[1]).abs().max()
    """)


def test_sharding3_world2_matches_single():
    """stage-3 sharded training on 2 ranks == single-process training."""
    run_dist("""
        import copy, torch
        import paddle_amd as paddle
        from paddle_amd.distributed.fleet.sharding import GroupShardedStage3, ShardedAdamW
        from paddle_amd.models import build_gpt, GPTPretrainingCriterion
        paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        paddle.seed(0)
        torch.manual_seed(0)
        m_ref = build_gpt('gpt3-tiny')
        m = copy.deepcopy(m_ref)
        loss_fn = GPTPretrainingCriterion()
        torch.manual_seed(123)
        ids_full = torch.randint(0, 1024, (4, 64))
        # single-process reference on the FULL batch
        opt_ref = paddle.optimizer.AdamW(learning_rate=1e-3, parameters=m_ref.parameters(),
                                         weight_decay=0.01)
        for _ in range(3):
            l_ref = loss_fn(m_ref(ids_full), ids_full)
            l_ref.backward()
            opt_ref.step(); opt_ref.clear_grad()
        # sharded DP: each rank takes half the batch
        ids = ids_full[r * 2:(r + 1) * 2]
        w = GroupShardedStage3(m)
        opt = ShardedAdamW(w, learning_rate=1e-3, weight_decay=0.01)
        for _ in range(3):
            # per-rank mean loss; grads averaged by reduce-scatter -> == full-batch mean
            l = loss_fn(w(ids), ids)
            l.backward()
            opt.step(); opt.clear_grad()
        # compare full params vs reference
        w.get_all_parameters()
        for (n1, p1), (n2, p2) in zip(w.named_parameters(), m_ref.named_parameters()):
            assert torch.allclose(p1, p2, rtol=2e-3, atol=2e-3), (n1, (p1-p2).abs().max())
        print('rank', r, 'ok')
    """, timeout=360)


def test_fleet_init_topology():
    run_dist("""
        import paddle_amd as paddle
        from paddle_amd.distributed import fleet
        strategy = fleet.DistributedStrategy()
        strategy.hybrid_configs = {"dp_degree": 2, "mp_degree": 1, "pp_degree": 1,
                                   "sharding_degree": 1}
        fleet.init(is_collective=True, strategy=strategy)
        hcg = fleet.get_hybrid_communicate_group()
        assert hcg.get_data_parallel_world_size() == 2
        assert hcg.get_model_parallel_world_size() == 1
        r = paddle.distributed.get_rank()
        assert hcg.get_data_parallel_rank() == r
    """)


def test_vocab_parallel_ce_gloo():
    """TP vocab-parallel cross entropy == plain CE (2-way vocab split)."""
    run_dist("""
        import torch
        import paddle_amd as paddle
        from paddle_amd.distributed import fleet
        from paddle_amd.distributed.fleet.mpu import ParallelCrossEntropy
        strategy = fleet.DistributedStrategy()
        strategy.hybrid_configs = {"dp_degree": 1, "mp_degree": 2, "pp_degree": 1,
                                   "sharding_degree": 1}
        fleet.init(is_collective=True, strategy=strategy)
        r = paddle.distributed.get_rank()
        torch.manual_seed(5)
        logits = torch.randn(6, 20)
        labels = torch.randint(0, 20, (6,))
        local = logits[:, r * 10:(r + 1) * 10].clone().requires_grad_(True)
        ce = ParallelCrossEntropy()
        loss = ce(local, labels).mean()
        ref = torch.nn.functional.cross_entropy(logits, labels)
        assert torch.allclose(loss, ref, atol=1e-5), (loss, ref)
        loss.backward()
        full = logits.clone().requires_grad_(True)
        torch.nn.functional.cross_entropy(full, labels).backward()
        assert torch.allclose(local.grad, full.grad[:, r*10:(r+1)*10], atol=1e-5)
    """)


def test_sharded_checkpoint_reshard_w2_to_w1(tmp_path):
    """Save the ZeRO-3 optimizer state on 2 ranks, restore on 1 rank:
    load_state_dict reassembles each unit's flat buffers from the
    overlapping saved slices (reference: distributed/checkpoint/
    load_state_dict.py resharding)."""
    ckpt = str(tmp_path / "distcp")
    run_dist(f"""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                           ShardedAdamW)

        torch.manual_seed(0)   # same init on both ranks (sync_init also covers it)
        m = paddle.nn.Sequential(paddle.nn.Linear(16, 32), paddle.nn.Linear(32, 16))
        wrapped = GroupShardedStage3(m, device=torch.device("cpu"))
        opt = ShardedAdamW(wrapped, learning_rate=1e-2)
        x = torch.randn(4, 16)
        loss = wrapped(x).pow(2).mean()
        loss.backward()
        opt.step()
        opt.clear_grad()
        opt.save_sharded({ckpt!r})
    """, world_size=2)

    # restore single-process: world 1 shard == the full flat buffer
    import torch
    import paddle_amd as paddle
    from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                       ShardedAdamW)
    torch.manual_seed(0)
    m = paddle.nn.Sequential(paddle.nn.Linear(16, 32), paddle.nn.Linear(32, 16))
    wrapped = GroupShardedStage3(m, device=torch.device("cpu"))
    opt = ShardedAdamW(wrapped, learning_rate=1e-2)
    before = [u.master.clone() for u in wrapped._units]
    opt.load_sharded(ckpt)
    assert opt.step_count == 1
    changed = any(not torch.equal(b, u.master)
                  for b, u in zip(before, wrapped._units))
    assert changed, "restore did not modify masters"
    # the restored single-rank state must equal re-running the same step
    # locally (identical init + same grads => identical AdamW result)
    torch.manual_seed(0)
    m2 = paddle.nn.Sequential(paddle.nn.Linear(16, 32), paddle.nn.Linear(32, 16))
    w2 = GroupShardedStage3(m2, device=torch.device("cpu"))
    o2 = ShardedAdamW(w2, learning_rate=1e-2)
    x = torch.randn(4, 16)   # NOTE: ranks saw different x; see below
    # grads differ across ranks in the saved run (different torch.randn
    # per-rank after seed reset? both ranks seeded 0 -> same x), so the
    # single-process rerun with the same seed reproduces the same step.
    loss = w2(x).pow(2).mean()
    loss.backward()
    o2.step()
    for u_loaded, u_ref in zip(wrapped._units, w2._units):
        torch.testing.assert_close(u_loaded.master, u_ref.master,
                                   atol=1e-5, rtol=1e-5)


def test_sharding_stage1_stage2_match_single():
    """group_sharded_parallel levels os (stage-1) and os_g (stage-2) on two
    ranks produce the same updated params as single-process training
    (reference: sharding stage semantics, dygraph_sharding_optimizer.py /
    group_sharded stage2)."""
    run_dist("""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed.fleet.sharding import group_sharded_parallel
        r = paddle.distributed.get_rank()

        for level in ("os", "os_g"):
            torch.manual_seed(7)
            m = paddle.nn.Sequential(paddle.nn.Linear(8, 16),
                                     paddle.nn.Linear(16, 8))
            inner = paddle.optimizer.AdamW(learning_rate=1e-2,
                                           parameters=m.parameters())
            wrapped, opt, _ = group_sharded_parallel(m, inner, level)
            torch.manual_seed(11)
            x = torch.randn(4, 8)
            loss = wrapped(x).pow(2).mean()
            loss.backward()
            opt.step()
            opt.clear_grad()

            # single-process reference
            torch.manual_seed(7)
            m2 = paddle.nn.Sequential(paddle.nn.Linear(8, 16),
                                      paddle.nn.Linear(16, 8))
            o2 = paddle.optimizer.AdamW(learning_rate=1e-2,
                                        parameters=m2.parameters())
            torch.manual_seed(11)
            x2 = torch.randn(4, 8)
            l2 = m2(x2).pow(2).mean()
            l2.backward()
            o2.step()
            for a, b in zip(m.parameters(), m2.parameters()):
                torch.testing.assert_close(a, b, atol=2e-5, rtol=2e-5), level
        print("rank", r, "stage1/2 ok")
    """, world_size=2)


def test_sharding3_grad_accumulation_matches_single():
    """no_sync() micro-batch accumulation on 2 ranks equals single-process
    accumulation over the same 4 micro-batches (exercises the fp32
    grad-accumulate branch of reduce_grads)."""
    run_dist("""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                           ShardedAdamW)
        r = paddle.distributed.get_rank()
        torch.manual_seed(0)
        m = paddle.nn.Sequential(paddle.nn.Linear(8, 16), paddle.nn.Linear(16, 8))
        wrapped = GroupShardedStage3(m, device=torch.device("cpu"))
        opt = ShardedAdamW(wrapped, learning_rate=1e-2)
        torch.manual_seed(17)
        xs = [torch.randn(2, 8) for _ in range(4)]
        mine = xs[r * 2:(r + 1) * 2]          # 2 micro-batches per rank
        with wrapped.no_sync():
            loss = wrapped(mine[0]).pow(2).mean()
            loss.backward()
        # final micro-batch syncs (accumulate branch then reduce)
        loss = wrapped(mine[1]).pow(2).mean()
        loss.backward()
        opt.step()
        opt.clear_grad()

        # single-process reference: mean over ranks of mean over micro-steps
        torch.manual_seed(0)
        m2 = paddle.nn.Sequential(paddle.nn.Linear(8, 16), paddle.nn.Linear(16, 8))
        w2 = GroupShardedStage3(m2, device=torch.device("cpu"), group=None)
        o2 = ShardedAdamW(w2, learning_rate=1e-2)
        # emulate: rank grads = sum of its 2 micro losses' grads; then mean
        # over ranks -> equivalent single-process grad = mean over ranks of
        # (sum over micro); build it by scaling
        for i, x in enumerate(xs):
            loss = w2(x).pow(2).mean() / 2.0   # 2 ranks averaged
            loss.backward()
        o2.step()
        for a, b in zip(wrapped._units, w2._units):
            torch.testing.assert_close(a.master, b.master, atol=2e-5, rtol=2e-5)
        print("rank", r, "accum ok")
    """, world_size=2)


def test_sharding3_main_grad_fp32_accumulation():
    """use_main_grad=True accumulates micro-grads in fp32 and matches the
    single-process result (reference: mixed_precision_utils main_grad)."""
    run_dist("""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                           ShardedAdamW)
        r = paddle.distributed.get_rank()
        torch.manual_seed(0)
        m = paddle.nn.Sequential(paddle.nn.Linear(8, 16), paddle.nn.Linear(16, 8))
        wrapped = GroupShardedStage3(m, device=torch.device("cpu"),
                                     use_main_grad=True)
        opt = ShardedAdamW(wrapped, learning_rate=1e-2)
        torch.manual_seed(23)
        xs = [torch.randn(2, 8) for _ in range(4)]
        mine = xs[r * 2:(r + 1) * 2]
        with wrapped.no_sync():
            wrapped(mine[0]).pow(2).mean().backward()
        wrapped(mine[1]).pow(2).mean().backward()
        opt.step()
        opt.clear_grad()

        torch.manual_seed(0)
        m2 = paddle.nn.Sequential(paddle.nn.Linear(8, 16), paddle.nn.Linear(16, 8))
        w2 = GroupShardedStage3(m2, device=torch.device("cpu"))
        o2 = ShardedAdamW(w2, learning_rate=1e-2)
        for x in xs:
            (w2(x).pow(2).mean() / 2.0).backward()
        o2.step()
        for a, b in zip(wrapped._units, w2._units):
            torch.testing.assert_close(a.master, b.master, atol=2e-5, rtol=2e-5)
        print("rank", r, "main_grad ok")
    """, world_size=2)


def test_merge_sharded_checkpoint(tmp_path):
    """Offline merge of a 2-rank sharded save rebuilds full flat tensors."""
    ckpt = str(tmp_path / "m")
    run_dist(f"""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                           ShardedAdamW)
        torch.manual_seed(0)
        m = paddle.nn.Linear(16, 16)
        w = GroupShardedStage3(m, device=torch.device("cpu"))
        opt = ShardedAdamW(w, learning_rate=1e-2)
        w(torch.randn(2, 16)).sum().backward()
        opt.step()
        opt.save_sharded({ckpt!r})
    """, world_size=2)
    from paddle_amd.distributed.checkpoint import merge_sharded
    full = merge_sharded(ckpt)
    import torch
    masters = [v for k, v in full.items() if k.endswith("_master")]
    assert masters and all(v.dim() == 1 for v in masters)
    total = sum(v.numel() for v in masters)
    assert total >= 16 * 16 + 16          # weight + bias (padded)
    assert full["step"] == 1


def test_distributed_longtail_api(tmp_path):
    """split/ParallelMode/DistModel/to_static/unshard surface (reference
    distributed/__init__.py exports)."""
    import torch
    import paddle_amd as paddle
    d = paddle.distributed
    assert d.ParallelMode.TENSOR_PARALLEL == 1
    assert d.ReduceType.kRedSum == 0
    # split at world 1 == plain ops
    x = torch.randn(3, 8)
    out = d.split(x, (8, 6), operation="linear", axis=1, num_partitions=1)
    assert out.shape == (3, 6)
    emb = d.split(torch.tensor([[1, 2]]), (16, 8), operation="embedding")
    assert emb.shape == (1, 2, 8)
    # DistModel one train step
    net = torch.nn.Linear(4, 2)
    opt = paddle.optimizer.SGD(learning_rate=0.1, parameters=net.parameters())
    dm = d.to_static(net, loss=torch.nn.functional.mse_loss, optimizer=opt)
    w0 = net.weight.detach().clone()
    loss = dm(torch.randn(5, 4), torch.randn(5, 2))
    assert loss.dim() == 0 and not torch.allclose(net.weight, w0)
    dm.eval()
    _ = dm(torch.randn(5, 4), torch.randn(5, 2))
    # unshard on a plain tensor is identity
    t = torch.randn(4)
    assert d.unshard_dtensor(t) is t
    # shard_dataloader wraps and iterates
    dl = [[torch.ones(2, 3), torch.zeros(2)]]
    from paddle_amd.distributed.auto_parallel import ProcessMesh
    sd = d.shard_dataloader(dl, ProcessMesh([[0]] if False else [0]))
    batches = list(sd)
    assert len(batches) == 1 and batches[0][0].shape == (2, 3)


def test_sharded_checkpoint_reshard_w1_to_w2(tmp_path):
    """Inverse resharding: save ZeRO-3 state at world 1, restore on 2
    ranks -- each rank's shard must be the matching slice of the
    single-rank flat state."""
    import torch
    import paddle_amd as paddle
    from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                       ShardedAdamW)
    ckpt = str(tmp_path / "distcp")
    torch.manual_seed(0)
    m = paddle.nn.Sequential(paddle.nn.Linear(16, 32), paddle.nn.Linear(32, 16))
    wrapped = GroupShardedStage3(m, device=torch.device("cpu"))
    opt = ShardedAdamW(wrapped, learning_rate=1e-2)
    x = torch.randn(4, 16)
    loss = wrapped(x).pow(2).mean()
    loss.backward()
    opt.step()
    opt.clear_grad()
    opt.save_sharded(ckpt)
    ref = str(tmp_path / "ref.pt")
    torch.save([u.master.clone() for u in wrapped._units], ref)

    run_dist(f"""
        import torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        from paddle_amd.distributed.fleet.sharding import (GroupShardedStage3,
                                                           ShardedAdamW)
        torch.manual_seed(0)
        m = paddle.nn.Sequential(paddle.nn.Linear(16, 32), paddle.nn.Linear(32, 16))
        wrapped = GroupShardedStage3(m, device=torch.device("cpu"))
        opt = ShardedAdamW(wrapped, learning_rate=1e-2)
        opt.load_sharded({ckpt!r})
        assert opt.step_count == 1
        full = torch.load({ref!r}, weights_only=False)
        r = paddle.distributed.get_rank()
        for u, f in zip(wrapped._units, full):
            n = u.shard_size
            want = f[r * n:(r + 1) * n]
            torch.testing.assert_close(u.master[:len(want)], want,
                                       atol=1e-6, rtol=1e-6)
        print("rank", r, "w1->w2 reshard ok")
    """, world_size=2)
