"""Sequence-parallel and context-parallel correctness on gloo (2 ranks)."""
import pytest

from dist_util import run_dist


def test_sp_ops_roundtrip_and_linears():
    run_dist("""
        import torch
        import paddle_amd as paddle
        from paddle_amd.distributed import fleet
        strategy = fleet.DistributedStrategy()
        strategy.hybrid_configs = {"dp_degree": 1, "mp_degree": 2, "pp_degree": 1,
                                   "sharding_degree": 1}
        fleet.init(is_collective=True, strategy=strategy)
        from paddle_amd.distributed.fleet.sequence_parallel import (
            AllGatherOp, ReduceScatterOp, ColumnSequenceParallelLinear,
            RowSequenceParallelLinear, scatter, gather)
        r = paddle.distributed.get_rank()
        g = fleet.get_hybrid_communicate_group().get_model_parallel_group()
        torch.manual_seed(0)
        full = torch.randn(8, 4)          # [seq, h]
        mine = scatter(full, g)
        assert mine.shape == (4, 4)
        assert torch.allclose(mine, full[r*4:(r+1)*4])
        back = gather(mine, g)
        assert torch.allclose(back, full)
        # allgather fwd == gather; reduce-scatter fwd sums + splits
        ag = AllGatherOp.apply(mine, g)
        assert torch.allclose(ag, full)
        rs = ReduceScatterOp.apply(full, g)   # both ranks same full -> 2x slice
        assert torch.allclose(rs, 2 * full[r*4:(r+1)*4])

        # col-SP + row-SP == plain mlp on the full sequence
        torch.manual_seed(1)
        W1 = torch.randn(4, 8)
        W2 = torch.randn(8, 4)
        x_full = torch.randn(8, 4)
        ref = torch.relu(x_full @ W1) @ W2
        col = ColumnSequenceParallelLinear(4, 8, has_bias=False, mp_group=g)
        row = RowSequenceParallelLinear(8, 4, has_bias=False, mp_group=g)
        with torch.no_grad():
            col.weight.copy_(W1[:, r*4:(r+1)*4])
            row.weight.copy_(W2[r*4:(r+1)*4, :])
        out_local = row(torch.relu(col(x_full[r*4:(r+1)*4])))
        assert torch.allclose(out_local, ref[r*4:(r+1)*4], atol=1e-5), \\
            (out_local - ref[r*4:(r+1)*4]).abs().max()
        print("rank", r, "sp ok")
    """)


def test_ulysses_attention_matches_single():
    run_dist("""
        import math, torch
        import paddle_amd as paddle
        paddle.distributed.init_parallel_env()
        r = paddle.distributed.get_rank()
        g = paddle.distributed.new_group([0, 1])
        from paddle_amd.distributed.fleet.context_parallel import ulysses_attention
        from paddle_amd.ops.functional import _sdpa_ref
        torch.manual_seed(3)
        b, s, h, d = 2, 16, 4, 16
        q = torch.randn(b, s, h, d)
        k = torch.randn(b, s, h, d)
        v = torch.randn(b, s, h, d)
        # reference on full seq (bhsd layout for _sdpa_ref)
        ref, _ = _sdpa_ref(q.permute(0,2,1,3), k.permute(0,2,1,3),
                           v.permute(0,2,1,3), 1/math.sqrt(d), True)
        ref = ref.permute(0, 2, 1, 3)
        sl = s // 2
        out = ulysses_attention(q[:, r*sl:(r+1)*sl].contiguous(),
                                k[:, r*sl:(r+1)*sl].contiguous(),
                                v[:, r*sl:(r+1)*sl].contiguous(),
                                cp_group=g, causal=True)
        assert torch.allclose(out, ref[:, r*sl:(r+1)*sl], atol=1e-5), \\
            (out - ref[:, r*sl:(r+1)*sl]).abs().max()
        # grads flow
        q2 = q[:, r*sl:(r+1)*sl].clone().requires_grad_(True)
        out2 = ulysses_attention(q2, k[:, r*sl:(r+1)*sl].contiguous(),
                                 v[:, r*sl:(r+1)*sl].contiguous(), cp_group=g,
                                 causal=True)
        out2.sum().backward()
        assert q2.grad is not None
        print("rank", r, "ulysses ok")
    """)


def test_ring_attention_matches_single():
    run_dist("""
        import math
        import torch
        import paddle_amd as paddle
        from paddle_amd.distributed import fleet
        strategy = fleet.DistributedStrategy()
        strategy.hybrid_configs = {"dp_degree": 1, "mp_degree": 2, "pp_degree": 1,
                                   "sharding_degree": 1}
        fleet.init(is_collective=True, strategy=strategy)
        from paddle_amd.distributed.fleet.context_parallel import (
            ring_attention, zigzag_split, zigzag_merge)
        from paddle_amd.ops.functional import flash_attention
        g = fleet.get_hybrid_communicate_group().get_model_parallel_group()
        r, w = g.rank, g.nranks
        torch.manual_seed(7)
        b, s, h, d = 2, 32, 3, 16
        q = torch.randn(b, s, h, d)
        k = torch.randn(b, s, h, d)
        v = torch.randn(b, s, h, d)
        for causal in (True, False):
            # single-process reference fwd+bwd
            qr = q.clone().requires_grad_(True)
            kr = k.clone().requires_grad_(True)
            vr = v.clone().requires_grad_(True)
            ref, _ = flash_attention(qr, kr, vr, causal=causal)
            gout = torch.randn_like(ref)
            ref.backward(gout)
            # ring on zigzag shards
            ql = zigzag_split(q, w, r).requires_grad_(True)
            kl = zigzag_split(k, w, r).requires_grad_(True)
            vl = zigzag_split(v, w, r).requires_grad_(True)
            out, lse = ring_attention(ql, kl, vl, cp_group=g, causal=causal)
            out.backward(zigzag_split(gout, w, r))
            ref_l = zigzag_split(ref.detach(), w, r)
            assert torch.allclose(out, ref_l, atol=2e-3, rtol=2e-3), \\
                (causal, (out - ref_l).abs().max())
            for got, refg in ((ql.grad, qr.grad), (kl.grad, kr.grad),
                              (vl.grad, vr.grad)):
                want = zigzag_split(refg, w, r)
                assert torch.allclose(got, want, atol=5e-3, rtol=5e-3), \\
                    (causal, (got - want).abs().max())
        # zigzag_merge reassembles
        full = torch.arange(2 * w * 3.0).reshape(1, 2 * w * 3, 1, 1)
        shards = [zigzag_split(full, w, i) for i in range(w)]
        assert torch.allclose(zigzag_merge(shards, w), full)
        print("rank", r, "ring ok")
    """)


def test_sp_inner_overlap_linear_parity():
    run_dist("""
        import torch
        import paddle_amd as paddle
        from paddle_amd.distributed import fleet
        strategy = fleet.DistributedStrategy()
        strategy.hybrid_configs = {"dp_degree": 1, "mp_degree": 2, "pp_degree": 1,
                                   "sharding_degree": 1}
        fleet.init(is_collective=True, strategy=strategy)
        from paddle_amd.distributed.fleet.sequence_parallel import (
            ColumnSequenceParallelLinear, SPInnerOverlapLinear)
        r = paddle.distributed.get_rank()
        g = fleet.get_hybrid_communicate_group().get_model_parallel_group()
        torch.manual_seed(3)
        W = torch.randn(4, 8)
        B = torch.randn(8)
        x_full = torch.randn(8, 4)

        col = ColumnSequenceParallelLinear(4, 8, has_bias=True, mp_group=g)
        ovl = SPInnerOverlapLinear(4, 8, has_bias=True, mp_group=g)
        with torch.no_grad():
            for m in (col, ovl):
                m.weight.copy_(W[:, r*4:(r+1)*4])
                m.bias.copy_(B[r*4:(r+1)*4])

        xs1 = x_full[r*4:(r+1)*4].clone().requires_grad_(True)
        xs2 = x_full[r*4:(r+1)*4].clone().requires_grad_(True)
        o1 = col(xs1)
        o2 = ovl(xs2)
        assert torch.allclose(o1, o2, atol=1e-5), (o1 - o2).abs().max()
        dy = torch.randn_like(o1)
        o1.backward(dy)
        o2.backward(dy)
        assert torch.allclose(xs1.grad, xs2.grad, atol=1e-5)
        assert torch.allclose(col.weight.grad, ovl.weight.grad, atol=1e-5)
        assert torch.allclose(col.bias.grad, ovl.bias.grad, atol=1e-5)
        print("rank", r, "sp-overlap ok")
    """)
