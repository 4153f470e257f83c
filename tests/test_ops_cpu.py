"""CPU multi-process collective tests (reference coverage model:
test/parallel/test_torch.py — every op x dtype x variant under a real
multi-process launch)."""
import pytest
import torch

from tests.parallel_util import run_workers


def test_allreduce_sum_np2():
    run_workers(2, """
        t = torch.arange(10, dtype=torch.float32) * (rank + 1)
        out = hvd.allreduce(t, average=False)
        expected = torch.arange(10, dtype=torch.float32) * 3
        assert torch.allclose(out, expected), (out, expected)
    """)


def test_allreduce_average_np3():
    run_workers(3, """
        t = torch.ones(7) * (rank + 1)
        out = hvd.allreduce(t)  # default Average
        assert torch.allclose(out, torch.full((7,), 2.0)), out
    """)


def test_allreduce_dtypes_np2():
    run_workers(2, """
        for dtype in [torch.float32, torch.float64, torch.int32, torch.int64,
                      torch.uint8, torch.int8, torch.float16, torch.bfloat16]:
            t = torch.arange(5).to(dtype)
            out = hvd.allreduce(t, average=False,
                                name=f"dt.{str(dtype)}")
            assert torch.allclose(out.float(), torch.arange(5).float() * 2), \
                (dtype, out)
    """)


def test_allreduce_min_max_product_np2():
    run_workers(2, """
        t = torch.tensor([1.0, 5.0]) if rank == 0 else torch.tensor([3.0, 2.0])
        mn = hvd.allreduce(t, op=hvd.Min, name="mn")
        mx = hvd.allreduce(t, op=hvd.Max, name="mx")
        pr = hvd.allreduce(t, op=hvd.Product, name="pr")
        assert torch.allclose(mn, torch.tensor([1.0, 2.0])), mn
        assert torch.allclose(mx, torch.tensor([3.0, 5.0])), mx
        assert torch.allclose(pr, torch.tensor([3.0, 10.0])), pr
    """)


def test_allreduce_prescale_postscale_np2():
    run_workers(2, """
        t = torch.ones(4)
        out = hvd.allreduce(t, average=False, prescale_factor=2.0,
                            postscale_factor=0.5, name="scaled")
        # (1*2 + 1*2) * 0.5 = 2
        assert torch.allclose(out, torch.full((4,), 2.0)), out
    """)


def test_allreduce_inplace_and_async_np2():
    run_workers(2, """
        t = torch.ones(3)
        hvd.allreduce_(t, average=False, name="inplace")
        assert torch.allclose(t, torch.full((3,), 2.0)), t
        h = hvd.allreduce_async(torch.ones(3), average=False, name="async1")
        assert hvd.synchronize(h).sum().item() == 6.0
        h2 = hvd.allreduce_async_(torch.ones(3), average=False, name="async2")
        while not hvd.poll(h2):
            pass
        assert hvd.synchronize(h2).sum().item() == 6.0
    """)


def test_grouped_allreduce_np2():
    run_workers(2, """
        ts = [torch.ones(5) * (rank + 1), torch.ones(3) * (rank + 10)]
        outs = hvd.grouped_allreduce(ts, average=False, name="grp")
        assert torch.allclose(outs[0], torch.full((5,), 3.0)), outs[0]
        assert torch.allclose(outs[1], torch.full((3,), 21.0)), outs[1]
    """)


def test_allgather_same_shape_np2():
    run_workers(2, """
        t = torch.arange(6).reshape(3, 2).float() * (rank + 1)
        out = hvd.allgather(t)
        assert out.shape == (6, 2), out.shape
        assert torch.allclose(out[:3], torch.arange(6).reshape(3, 2).float())
        assert torch.allclose(out[3:], torch.arange(6).reshape(3, 2).float()*2)
    """)


def test_allgather_variable_shape_np3():
    run_workers(3, """
        n = rank + 1
        t = torch.full((n, 2), float(rank))
        out = hvd.allgather(t, name="agv")
        assert out.shape == (6, 2), out.shape
        assert torch.allclose(out[0:1], torch.zeros(1, 2))
        assert torch.allclose(out[1:3], torch.ones(2, 2))
        assert torch.allclose(out[3:6], torch.full((3, 2), 2.0))
    """)


def test_broadcast_np2():
    run_workers(2, """
        t = torch.arange(5).float() * (rank + 1)
        out = hvd.broadcast(t, root_rank=1, name="bc")
        assert torch.allclose(out, torch.arange(5).float() * 2), out
        t2 = torch.full((3,), float(rank))
        hvd.broadcast_(t2, root_rank=0, name="bc2")
        assert torch.allclose(t2, torch.zeros(3)), t2
    """)


def test_alltoall_np2():
    run_workers(2, """
        # rank r sends rows [r*2, r*2+1) to each peer
        t = torch.arange(8).reshape(4, 2).float() + rank * 100
        out, rsplits = hvd.alltoall(t, splits=[2, 2], name="a2a")
        assert out.shape == (4, 2), out.shape
        if rank == 0:
            expected = torch.cat([torch.arange(4).reshape(2, 2).float(),
                                  torch.arange(4).reshape(2, 2).float() + 100])
        else:
            expected = torch.cat([torch.arange(4, 8).reshape(2, 2).float(),
                                  torch.arange(4, 8).reshape(2,2).float()+100])
        assert torch.allclose(out, expected), (out, expected)
        assert rsplits.tolist() == [2, 2]
    """)


def test_alltoall_uneven_np2():
    run_workers(2, """
        if rank == 0:
            t = torch.arange(3).float()
            splits = [1, 2]
        else:
            t = torch.arange(10, 13).float()
            splits = [2, 1]
        out, rsplits = hvd.alltoall(t, splits=splits, name="a2av")
        if rank == 0:
            assert out.tolist() == [0.0, 10.0, 11.0], out
            assert rsplits.tolist() == [1, 2]
        else:
            assert out.tolist() == [1.0, 2.0, 12.0], out
            assert rsplits.tolist() == [2, 1]
    """)


def test_reducescatter_np2():
    run_workers(2, """
        t = torch.arange(8).float().reshape(4, 2) * (rank + 1)
        out = hvd.reducescatter(t, op=hvd.Sum, name="rs")
        full = torch.arange(8).float().reshape(4, 2) * 3
        mine = full[rank * 2:(rank + 1) * 2]
        assert torch.allclose(out, mine), (out, mine)
    """)


def test_reducescatter_uneven_np2():
    run_workers(2, """
        t = torch.arange(5).float() * (rank + 1)  # first dim 5: uneven over 2
        out = hvd.reducescatter(t, op=hvd.Sum, name="rsv")
        full = torch.arange(5).float() * 3
        if rank == 0:
            assert torch.allclose(out, full[:3]), out
        else:
            assert torch.allclose(out, full[3:]), out
    """)


def test_barrier_np3():
    run_workers(3, """
        import time
        time.sleep(0.1 * rank)
        hvd.barrier()
    """)


def test_join_np2():
    run_workers(2, """
        # rank 0 runs 3 batches, rank 1 runs 1 batch; join pads the gap
        nb = 3 if rank == 0 else 1
        for i in range(nb):
            out = hvd.allreduce(torch.ones(4), average=False, name=f"b{i}")
        last = hvd.join()
        assert last == 0 or last == 1
    """, timeout=240)


def test_process_sets_np3():
    run_workers(3, """
        ps = hvd.add_process_set(hvd.ProcessSet([0, 2]))
        if rank in (0, 2):
            out = hvd.allreduce(torch.ones(3) * (rank + 1), average=False,
                                process_set=ps, name="sub")
            assert torch.allclose(out, torch.full((3,), 4.0)), out
        # global op still works for everyone
        out = hvd.allreduce(torch.ones(2), average=False, name="glob")
        assert out.sum().item() == 6.0
    """)


def test_error_mismatched_shape_np2():
    run_workers(2, """
        n = 4 if rank == 0 else 5
        try:
            hvd.allreduce(torch.ones(n), average=False, name="bad")
            raise SystemExit("expected an error for mismatched shapes")
        except RuntimeError as e:
            assert "Mismatched" in str(e), e
    """)


def test_steady_state_cache_np2():
    """Many iterations of the same tensors: exercises the response-cache
    fast path (steady-state training must not touch the coordinator)."""
    run_workers(2, """
        for i in range(50):
            out = hvd.allreduce(torch.ones(1000), average=False, name="steady")
            assert out.sum().item() == 2000.0
    """)


def test_parallel_grid_sets_np4():
    run_workers(4, """
        from horovod_amd.parallel import grid_process_sets
        my_tp, my_dp, tp_sets, dp_sets = grid_process_sets(2)
        # tp allreduce within my row
        out = hvd.allreduce(torch.ones(2) * (rank + 1), average=False,
                            process_set=my_tp, name="tp")
        row = rank // 2
        expected = (2 * row + 1) + (2 * row + 2)
        assert out[0].item() == expected, (out, expected)
        # dp allreduce within my column
        out = hvd.allreduce(torch.ones(2) * (rank + 1), average=False,
                            process_set=my_dp, name="dp")
        col = rank % 2
        expected = (col + 1) + (col + 3)
        assert out[0].item() == expected, (out, expected)
    """)


def test_noncontiguous_np2():
    run_workers(2, """
        base = torch.arange(12).float().reshape(3, 4)
        t = base.t()  # non-contiguous view
        out = hvd.allreduce(t, average=False, name="nc")
        assert torch.allclose(out, base.t() * 2), out
        try:
            hvd.allreduce_(t, average=False, name="nc2")
            raise SystemExit("expected error for in-place non-contiguous")
        except ValueError:
            pass
    """)


def test_subset_ops_nonmember_root_np3():
    """All collectives on a process set excluding global rank 0: the
    coordinator participates only as a relay (lock-step frames)."""
    run_workers(3, """
        ps = hvd.add_process_set(hvd.ProcessSet([1, 2]))
        if rank in (1, 2):
            li = [1, 2].index(rank)
            out = hvd.allreduce(torch.ones(4) * (rank), average=False,
                                process_set=ps, name="s_ar")
            assert out[0].item() == 3.0, out
            g = hvd.allgather(torch.full((2, 2), float(rank)),
                              process_set=ps, name="s_ag")
            assert g.shape == (4, 2) and g[0, 0] == 1.0 and g[2, 0] == 2.0, g
            b = hvd.broadcast(torch.ones(3) * rank, root_rank=2,
                              process_set=ps, name="s_bc")
            assert b[0].item() == 2.0, b
            o, rs = hvd.alltoall(torch.arange(4).float() + rank * 10,
                                 splits=[2, 2], process_set=ps, name="s_a2a")
            exp = ([10.0, 11.0, 20.0, 21.0] if rank == 1
                   else [12.0, 13.0, 22.0, 23.0])
            assert o.tolist() == exp, (o, exp)
            r = hvd.reducescatter(torch.arange(4).float(), op=hvd.Sum,
                                  process_set=ps, name="s_rs")
            exp = [0.0, 2.0] if rank == 1 else [4.0, 6.0]
            assert r.tolist() == exp, r
            hvd.barrier(process_set=ps)
        # everyone joins a final global barrier
        hvd.barrier()
    """)


def test_fusion_threshold_splitting_np2():
    """Many tensors with a tiny fusion threshold: multiple fused responses
    per cycle, all results exact."""
    run_workers(2, """
        ts = [torch.full((1000,), float(i + rank)) for i in range(40)]
        outs = hvd.grouped_allreduce(ts, average=False, name="ft")
        for i, o in enumerate(outs):
            assert torch.allclose(o, torch.full((1000,), 2.0 * i + 1.0)), i
    """, extra_env={"HOROVOD_FUSION_THRESHOLD": str(8 * 1024)})


def test_allgather_broadcast_dtypes_np2():
    run_workers(2, """
        for dtype in [torch.uint8, torch.int64, torch.float16, torch.bool,
                      torch.bfloat16]:
            t = (torch.arange(4) % 2).to(dtype)
            g = hvd.allgather(t, name=f"agd{dtype}")
            assert g.shape[0] == 8, (dtype, g.shape)
            b = hvd.broadcast(t, root_rank=0, name=f"bcd{dtype}")
            assert torch.equal(b, t), dtype
    """)


def test_remove_process_set_np2():
    run_workers(2, """
        ps = hvd.add_process_set(hvd.ProcessSet([0, 1]))
        out = hvd.allreduce(torch.ones(2), average=False, process_set=ps,
                            name="rps1")
        assert out.sum().item() == 4.0
        hvd.remove_process_set(ps)
        ps2 = hvd.add_process_set(hvd.ProcessSet([0, 1]))
        out = hvd.allreduce(torch.ones(2), average=False, process_set=ps2,
                            name="rps2")
        assert out.sum().item() == 4.0
    """)


def test_cache_invalidation_reshape_np2():
    """Same tensor name with a changed shape: the cache entry must be
    invalidated and renegotiated, not reused."""
    run_workers(2, """
        for shape in [(8,), (8,), (16,), (16,), (8,)]:
            out = hvd.allreduce(torch.ones(*shape), average=False,
                                name="reshape_me")
            assert out.sum().item() == 2.0 * shape[0], (shape, out.sum())
    """)


def test_grouped_allgather_reducescatter_np2():
    run_workers(2, """
        ts = [torch.full((2, 3), float(rank)), torch.full((1, 3), float(rank))]
        outs = hvd.grouped_allgather(ts, name="gag")
        assert outs[0].shape == (4, 3) and outs[1].shape == (2, 3)
        assert outs[0][0, 0] == 0.0 and outs[0][2, 0] == 1.0
        rs_in = [torch.ones(4, 2) * (rank + 1), torch.ones(2, 2) * (rank + 1)]
        routs = hvd.grouped_reducescatter(rs_in, op=hvd.Sum, name="grs")
        assert routs[0].shape == (2, 2) and routs[0][0, 0] == 3.0
        assert routs[1].shape == (1, 2) and routs[1][0, 0] == 3.0
    """)


def test_reducescatter_default_average_np2():
    run_workers(2, """
        t = torch.full((4,), 2.0 * (rank + 1))
        out = hvd.reducescatter(t, name="rsavg")  # default = Average
        assert torch.allclose(out, torch.full((2,), 3.0)), out
    """)


def test_scalar_and_edge_shapes_np2():
    run_workers(2, """
        # 0-dim tensor allreduce
        s = torch.tensor(float(rank + 1))
        out = hvd.allreduce(s, average=False, name="scalar")
        assert out.item() == 3.0
        # single-element broadcast
        b = torch.tensor([float(rank)])
        hvd.broadcast_(b, root_rank=1, name="sb")
        assert b.item() == 1.0
        # empty-ish allgather (one rank contributes zero rows)
        t = torch.ones(0 if rank == 0 else 2, 3)
        g = hvd.allgather(t, name="eag")
        assert g.shape == (2, 3), g.shape
    """)


def test_join_last_rank_np3():
    run_workers(3, """
        import time
        hvd.barrier()            # align ranks before the staggered delays
        time.sleep(0.7 * rank)   # rank 2 joins well after the others
        last = hvd.join()
        assert last == 2, last
    """)


def test_timeline_mark_cycles_np2(tmp_path):
    import json as _json
    tl = str(tmp_path / "tlm.json")
    run_workers(2, f"""
        hvd.start_timeline({tl!r})
        for i in range(5):
            hvd.allreduce(torch.ones(10), average=False, name=f"m{{i}}")
        hvd.stop_timeline()
    """, extra_env={"HOROVOD_TIMELINE_MARK_CYCLES": "1"})
    events = _json.load(open(tl))
    assert any(e.get("name", "").startswith("CYCLE_") for e in events)


def test_grouped_error_does_not_hang_np2():
    """One member of a group has mismatched shapes: all ranks must get an
    error (or complete), never hang."""
    run_workers(2, """
        ts = [torch.ones(4), torch.ones(6 if rank == 0 else 7)]
        try:
            hvd.grouped_allreduce(ts, average=False, name="poison")
            raise SystemExit("expected an error")
        except RuntimeError as e:
            assert "Mismatched" in str(e), e
    """, timeout=90)


def test_exhaustive_op_dtype_sweep_np2():
    """Reference-style exhaustive sweep: every reduce op x dtype x shape
    (test_torch.py's coverage model, compressed into one launch)."""
    run_workers(2, """
        import itertools
        ops = [("sum", hvd.Sum), ("min", hvd.Min), ("max", hvd.Max),
               ("prod", hvd.Product)]
        dtypes = [torch.float32, torch.float64, torch.float16,
                  torch.bfloat16, torch.int32, torch.int64, torch.uint8]
        shapes = [(1,), (17, 3), (128,)]
        for (opname, op), dtype, shape in itertools.product(ops, dtypes,
                                                            shapes):
            base = (torch.arange(int(torch.tensor(shape).prod()))
                    .reshape(shape) % 5 + 1)
            a = base.to(dtype)                  # rank 0 contribution
            b = (base * 2).clamp(max=9).to(dtype)  # rank 1 contribution
            mine = a if rank == 0 else b
            out = hvd.allreduce(mine, op=op,
                                name=f"sweep.{opname}.{dtype}.{len(shape)}")
            af, bf = a.double(), b.double()
            expected = {"sum": af + bf, "min": torch.minimum(af, bf),
                        "max": torch.maximum(af, bf),
                        "prod": af * bf}[opname]
            assert torch.allclose(out.double(), expected, rtol=1e-2,
                                  atol=1e-2), (opname, dtype, shape,
                                               out, expected)
    """, timeout=300)


def test_join_with_allgather_np2():
    run_workers(2, """
        if rank == 0:
            g = hvd.allgather(torch.full((2, 3), 7.0), name="jag")
            # rank 1 already joined: only our rows come back
            assert g.shape == (2, 3), g.shape
            out = hvd.allreduce(torch.ones(4), average=False, name="jar")
            assert out.sum().item() == 4.0  # peer contributes zeros
        hvd.join()
    """, timeout=120)


def test_integer_average_np2():
    """Integer Average floor-divides the sum (reference: DivideInPlace,
    mpi_ops_v2.cc:62-68) — including negative values."""
    run_workers(2, """
        t = torch.tensor([1, -1, 5, -5], dtype=torch.int64) * (rank + 1)
        out = hvd.allreduce(t, average=True, name="iavg")
        # sums: [3,-3,15,-15]; floor-div 2 -> [1,-2,7,-8]
        assert out.tolist() == [1, -2, 7, -8], out
    """)


def test_reducescatter_integer_average_np2():
    run_workers(2, """
        t = torch.tensor([1, -1, 5, -5], dtype=torch.int64) * (rank + 1)
        out = hvd.reducescatter(t, name="rsiavg")  # default Average
        full = [1, -2, 7, -8]  # floor(sum/2)
        mine = full[:2] if rank == 0 else full[2:]
        assert out.tolist() == mine, (out, mine)
    """)


def test_bool_allreduce_np2():
    run_workers(2, """
        t = torch.tensor([True, False, bool(rank)])
        out = hvd.allreduce(t, average=False, name="bool")
        assert out.tolist() == [True, False, True], out
    """)


def test_alltoall_cached_steady_state_np2():
    """Repeated alltoall with unchanged splits takes the cache fast path;
    changing splits invalidates and renegotiates correctly."""
    run_workers(2, """
        for i in range(30):
            t = torch.arange(4).float() + rank * 10 + i
            out, rs = hvd.alltoall(t, splits=[2, 2], name="a2ac")
            exp = ([0.0 + i, 1 + i, 10 + i, 11 + i] if rank == 0
                   else [2.0 + i, 3 + i, 12 + i, 13 + i])
            assert out.tolist() == exp, (i, out, exp)
        # changed splits -> signature invalidation -> still correct
        splits = [1, 3] if rank == 0 else [3, 1]
        t = torch.arange(4).float() + rank * 10
        out, rs = hvd.alltoall(t, splits=splits, name="a2ac")
        if rank == 0:
            assert out.tolist() == [0.0, 10.0, 11.0, 12.0], out
        else:
            assert out.tolist() == [1.0, 2.0, 3.0, 13.0], out
    """)


def test_remove_process_set_evicts_cache_np2():
    """Using a set (cached), removing it, then continuing global work must
    not produce phantom fast-path responses (busy-spin regression guard)."""
    run_workers(2, """
        import time
        ps = hvd.add_process_set([0, 1])
        for i in range(5):
            hvd.allreduce(torch.ones(4), average=False, process_set=ps,
                          name="cachedsub")
        hvd.remove_process_set(ps)
        for i in range(10):
            out = hvd.allreduce(torch.ones(2), average=False, name=f"g{i}")
            assert out.sum().item() == 4.0
        time.sleep(0.3)  # idle: bg loop must be able to sleep (no spin)
    """)


def test_cross_check_vs_torch_distributed_np2():
    """Independent oracle: every collective compared against
    torch.distributed (gloo) on the same random tensors."""
    import os as _os
    from horovod_amd.runner.launch import find_free_port
    port = find_free_port()
    run_workers(2, f"""
        import os
        import torch.distributed as dist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ["MASTER_PORT"] = "{port}"
        dist.init_process_group("gloo", rank=rank, world_size=size)
        torch.manual_seed(500 + rank)
        for trial in range(5):
            t = torch.randn(64, 3)
            # allreduce
            ours = hvd.allreduce(t, average=False, name=f"xc.ar.{{trial}}")
            ref = t.clone(); dist.all_reduce(ref)
            assert torch.allclose(ours, ref, atol=1e-6), "allreduce"
            # allgather
            ours = hvd.allgather(t, name=f"xc.ag.{{trial}}")
            outs = [torch.empty_like(t) for _ in range(size)]
            dist.all_gather(outs, t)
            assert torch.allclose(ours, torch.cat(outs), atol=1e-6), "ag"
            # broadcast
            b = t.clone(); ours = hvd.broadcast(b, root_rank=1,
                                                name=f"xc.bc.{{trial}}")
            ref = t.clone(); dist.broadcast(ref, src=1)
            assert torch.allclose(ours, ref, atol=1e-6), "bcast"
            # reducescatter (sum)
            ours = hvd.reducescatter(t, op=hvd.Sum, name=f"xc.rs.{{trial}}")
            full = t.clone(); dist.all_reduce(full)
            assert torch.allclose(ours, full[rank*32:(rank+1)*32],
                                  atol=1e-6), "rs"
            # alltoall (gloo lacks all_to_all: oracle via all_gather)
            ours, _ = hvd.alltoall(t, splits=[32, 32],
                                   name=f"xc.a2a.{{trial}}")
            everyone = [torch.empty_like(t) for _ in range(size)]
            dist.all_gather(everyone, t)
            expected = torch.cat([e.chunk(2)[rank] for e in everyone])
            assert torch.allclose(ours, expected, atol=1e-6), "a2a"
        dist.destroy_process_group()
    """, timeout=300)


def test_alltoall_int64_np2():
    run_workers(2, """
        t = torch.arange(4, dtype=torch.int64) + rank * 100
        out, rs = hvd.alltoall(t, splits=[2, 2], name="a2ai")
        exp = [0, 1, 100, 101] if rank == 0 else [2, 3, 102, 103]
        assert out.tolist() == exp, out
    """)


def test_concurrent_sets_grouped_compressed_np4():
    """Two overlapping process sets with grouped + wire-compressed ops
    interleaved with global collectives."""
    run_workers(4, """
        from horovod_amd.torch.compression import Compression
        even = hvd.add_process_set([0, 2])
        odd = hvd.add_process_set([1, 3])
        mine = even if rank % 2 == 0 else odd
        peer_sum = (rank % 2 + 1) + (rank % 2 + 3)  # ranks+1 summed in my set
        for i in range(10):
            ts = [torch.ones(100) * (rank + 1), torch.ones(50) * (rank + 1)]
            from horovod_amd.torch.compression import Compression as _C
            comp = _C.fp16 if i % 2 else None
            outs = hvd.grouped_allreduce(ts, average=False, process_set=mine,
                                         name=f"cs{i}{bool(comp)}",
                                         compression=comp)
            assert outs[0][0].item() == peer_sum, (i, outs[0][0])
            g = hvd.allreduce(torch.ones(10) * (rank + 1), average=False,
                              name=f"cg{i}", compression=Compression.fp16)
            assert abs(g[0].item() - 10.0) < 0.1, g[0]
        hvd.barrier()
    """)


def test_mixed_reduce_ops_not_fused_np2():
    """Round-2 regression (ADVICE high): concurrent allreduces with the same
    dtype/shape but different reduce ops must not fuse into one buffer."""
    run_workers(2, """
        a = torch.tensor([1.0, 5.0]) if rank == 0 else torch.tensor([3.0, 2.0])
        b = torch.tensor([2.0, 7.0]) if rank == 0 else torch.tensor([4.0, 1.0])
        hs = [hvd.allreduce_async(a, op=hvd.Sum, name="mix.sum"),
              hvd.allreduce_async(b, op=hvd.Max, name="mix.max"),
              hvd.allreduce_async(a, op=hvd.Min, name="mix.min"),
              hvd.allreduce_async(b, op=hvd.Product, name="mix.prod")]
        s, mx, mn, pr = [hvd.synchronize(h) for h in hs]
        assert torch.allclose(s, torch.tensor([4.0, 7.0])), s
        assert torch.allclose(mx, torch.tensor([4.0, 7.0])), mx
        assert torch.allclose(mn, torch.tensor([1.0, 2.0])), mn
        assert torch.allclose(pr, torch.tensor([8.0, 7.0])), pr
    """)


def test_minmax_steady_state_cache_np2():
    """Round-2 regression (ADVICE low): non-SUM ops must survive the cache
    fast path — same name repeated across many cycles."""
    run_workers(2, """
        for i in range(12):
            t = torch.tensor([float(rank + 1), float(10 - rank)])
            mn = hvd.allreduce(t, op=hvd.Min, name="steadymin")
            mx = hvd.allreduce(t, op=hvd.Max, name="steadymax")
            assert torch.allclose(mn, torch.tensor([1.0, 9.0])), (i, mn)
            assert torch.allclose(mx, torch.tensor([2.0, 10.0])), (i, mx)
    """)


def test_synchronize_fires_missing_hooks_np2():
    """Round-2 regression (ADVICE medium): per-rank conditional execution —
    a param whose hook never fired on one rank must still be allreduced by
    synchronize(), or peers deadlock."""
    run_workers(2, """
        m = torch.nn.Linear(4, 2, bias=True)
        with torch.no_grad():
            m.weight.fill_(1.0); m.bias.fill_(0.0)
        opt = torch.optim.SGD(m.parameters(), lr=0.0)
        opt = hvd.DistributedOptimizer(opt,
                                       named_parameters=m.named_parameters())
        x = torch.ones(1, 4)
        if rank == 0:
            loss = m(x).sum()          # both weight and bias get grads
        else:
            loss = (m.weight @ x.t()).sum()  # bias hook never fires
        loss.backward()
        opt.step()                      # must not hang; fires bias on rank 1
        # bias grad: rank0 contributed ones, rank1 zeros -> Average = 0.5
        assert torch.allclose(m.bias.grad, torch.full((2,), 0.5)), m.bias.grad
    """)


def test_adasum_set_backward_passes_np2():
    """Round-2 regression (ADVICE low): no AttributeError on the Adasum
    delta optimizer's set_backward_passes_per_step."""
    run_workers(2, """
        m = torch.nn.Linear(3, 3)
        opt = torch.optim.SGD(m.parameters(), lr=0.1)
        opt = hvd.DistributedOptimizer(opt, op=hvd.Adasum,
                                       named_parameters=m.named_parameters())
        opt.set_backward_passes_per_step(2)
        assert opt.backward_passes_per_step == 2
    """)


def test_broadcast_fresh_optimizer_state_np2():
    """Round-2: broadcast_optimizer_state on a FRESH optimizer initializes
    identical momentum state everywhere (reference functions.py:90-109)."""
    run_workers(2, """
        torch.manual_seed(100 + rank)  # different params per rank
        m = torch.nn.Linear(4, 4)
        opt = torch.optim.SGD(m.parameters(), lr=0.1, momentum=0.9)
        hvd.broadcast_parameters(m.state_dict(), root_rank=0)
        hvd.broadcast_optimizer_state(opt, root_rank=0, model=m)
        sd = opt.state_dict()
        assert len(sd['state']) == 2, sd['state'].keys()
        for s in sd['state'].values():
            assert 'momentum_buffer' in s
        # buffers must agree across ranks (both rebuilt from root)
        flat = torch.cat([s['momentum_buffer'].flatten()
                          for s in sd['state'].values()])
        mx = hvd.allreduce(flat, op=hvd.Max, name="mbmax")
        mn = hvd.allreduce(flat, op=hvd.Min, name="mbmin")
        assert torch.equal(mx, mn)
    """)


def test_reducescatter_prescale_np2():
    """Round-2 regression (ADVICE medium, CPU twin of the GPU fix):
    reducescatter honors prescale_factor."""
    run_workers(2, """
        t = torch.ones(4, 3)
        out = hvd.reducescatter(t, op=hvd.Sum, prescale_factor=3.0,
                                name="rsps")
        # 3*1 + 3*1 = 6 per element, 2 rows per rank
        assert out.shape == (2, 3), out.shape
        assert torch.allclose(out, torch.full((2, 3), 6.0)), out
    """)


def test_allgather_grad_uneven_np2():
    """Round-2: allgather autograd backward slices at the offset captured in
    forward (no per-backward dims collective) — uneven first dims."""
    run_workers(2, """
        n = 2 if rank == 0 else 3
        t = torch.full((n, 2), float(rank + 1), requires_grad=True)
        out = hvd.allgather(t, name="aggrad")
        assert out.shape == (5, 2)
        # d(loss)/dt where loss weights this rank's slice by (rank+2):
        w = torch.ones_like(out)
        w[(0 if rank == 0 else 2):(2 if rank == 0 else 5)] = 0.0
        loss = (out * w).sum()
        loss.backward()
        # grad of summed-other-slices: other rank's weight contributes 1 via
        # allreduce-sum of grad_output then slicing our rows
        assert t.grad.shape == t.shape
        assert torch.allclose(t.grad, torch.ones_like(t)), t.grad
    """)


def test_join_with_reducescatter_np2():
    """Round-2: joined ranks zero-substitute for reducescatter too
    (reference tensor_queue.cc:125-141 substitutes for every op type)."""
    run_workers(2, """
        if rank == 0:
            out = hvd.reducescatter(torch.full((4, 3), 2.0), op=hvd.Sum,
                                    name="jrs")
            # peer joined -> contributes zeros; I get rows 0..1
            assert out.shape == (2, 3), out.shape
            assert torch.allclose(out, torch.full((2, 3), 2.0)), out
        hvd.join()
    """, timeout=120)


def test_join_with_alltoall_np2():
    """Round-2: a joined rank posts zero-row sends but still receives what
    peers address to it (its split-matrix row is zeros)."""
    run_workers(2, """
        if rank == 0:
            t = torch.arange(12, dtype=torch.float32).reshape(4, 3)
            out, rs = hvd.alltoall(t, splits=torch.tensor([2, 2]),
                                   name="jata")
            # joined peer sent nothing; we get only our own first 2 rows
            assert out.shape == (2, 3), out.shape
            assert torch.allclose(out, t[:2]), out
            assert rs.tolist() == [2, 0], rs
        hvd.join()
    """, timeout=120)


def test_response_stream_hash_check_np3():
    """Round-2 (verdict weak 5): with HOROVOD_CHECK_RESPONSE_STREAM=1 every
    cycle cross-checks a running hash of executed responses; identical
    streams must pass through steady-state caching, groups and subsets."""
    run_workers(3, """
        ps = hvd.add_process_set(hvd.ProcessSet([0, 2]))
        for i in range(15):
            out = hvd.allreduce(torch.ones(64) * (rank + 1), average=False,
                                name="hsh")
            assert out[0].item() == 6.0
            if rank in (0, 2):
                hvd.allreduce(torch.ones(8), average=False, name="hsub",
                              process_set=ps)
            ts = [torch.ones(16) for _ in range(2)]
            hvd.grouped_allreduce(ts, average=False, name=f"hg{i % 3}")
    """, extra_env={"HOROVOD_CHECK_RESPONSE_STREAM": "1"}, timeout=240)


def test_mesh_large_transfers_np2():
    """Round-2 regression: ring/pairwise transfers far larger than the TCP
    socket queues (~10 MB) must not deadlock — blocking send() on Linux
    queues the whole request before returning, so the mesh sockets have to
    be non-blocking for the full-duplex poll loop to keep draining."""
    run_workers(2, """
        # 64 MB fp32 allreduce -> 32 MB ring segments each way
        n = 16 * 1024 * 1024
        t = torch.full((n,), float(rank + 1))
        out = hvd.allreduce(t, average=False, name="big.ar")
        assert out[0].item() == 3.0 and out[-1].item() == 3.0
        # 32 MB allgather (pairwise exchange)
        g = hvd.allgather(torch.full((n // 4, 2), float(rank)),
                          name="big.ag")
        assert g.shape[0] == n // 2
        # 48 MB broadcast (binomial: pure one-directional send)
        b = hvd.broadcast(torch.full((12 * 1024 * 1024,), 7.0),
                          root_rank=0, name="big.bc")
        assert b[-1].item() == 7.0
    """, timeout=420)


def test_subset_adasum_np4():
    """Round-2: CPU Adasum runs distributed VHDD over mesh links entirely
    within the subset's members (scalar sums no longer need the star, so
    subset Adasum works with non-members running free)."""
    run_workers(4, """
        ps = hvd.add_process_set(hvd.ProcessSet([1, 2]))
        if rank in (1, 2):
            def vec(r):
                return torch.cos(torch.arange(37, dtype=torch.float64) *
                                 (r + 2))
            def combine(a, b):
                dot = (a*b).sum(); na = (a*a).sum(); nb = (b*b).sum()
                return a*(1 - dot/(2*na)) + b*(1 - dot/(2*nb))
            expected = combine(vec(1), vec(2)).float()
            out = hvd.allreduce(vec(rank).float(), op=hvd.Adasum,
                                name="sada", process_set=ps)
            assert torch.allclose(out, expected, atol=1e-6), \
                (out - expected).abs().max()
        # everyone still agrees on a global op afterwards
        g = hvd.allreduce(torch.ones(4), average=False, name="after")
        assert g[0].item() == 4.0
    """, timeout=240)


def test_error_mismatched_dtype_np2():
    """Mismatched dtypes across ranks produce the reference's named error
    on every rank (reference controller.cc:599-602 error text)."""
    run_workers(2, """
        t = torch.ones(4) if rank == 0 else torch.ones(4, dtype=torch.float64)
        try:
            hvd.allreduce(t, average=False, name="baddt")
            raise SystemExit("expected a dtype-mismatch error")
        except RuntimeError as e:
            assert "Mismatched data types" in str(e), e
        # the job survives the error: a good op still completes
        out = hvd.allreduce(torch.ones(4), average=False, name="gooddt")
        assert out[0].item() == 2.0
    """)


def test_error_mismatched_root_np2():
    run_workers(2, """
        try:
            hvd.broadcast(torch.ones(4), root_rank=rank, name="badroot")
            raise SystemExit("expected a root-mismatch error")
        except RuntimeError as e:
            assert "root rank" in str(e).lower(), e
    """)


def test_error_mismatched_reduce_op_np2():
    run_workers(2, """
        op = hvd.Sum if rank == 0 else hvd.Max
        try:
            hvd.allreduce(torch.ones(4), op=op, name="badop")
            raise SystemExit("expected a reduce-op-mismatch error")
        except RuntimeError as e:
            assert "reduction op" in str(e).lower() or \
                "Mismatched" in str(e), e
    """)


def test_join_with_cached_steady_state_np2():
    """Round-2 regression: join() while the PEER's allreduces are in the
    cache fast path (same tensor name every step — the DistributedOptimizer
    pattern).  A joined rank must vote ready for every cache slot
    (reference controller.cc:130-134) or the peer's cached tensors never
    fire and both ranks deadlock."""
    run_workers(2, """
        nb = 12 if rank == 0 else 5
        for i in range(nb):
            # SAME name each iteration -> steady-state cache fast path
            out = hvd.allreduce(torch.ones(64), average=False, name="jcs")
            live = 2 if i < 5 else 1
            assert out[0].item() == float(live), (i, out[0].item())
        hvd.join()
        # both ranks usable afterwards
        g = hvd.allreduce(torch.ones(4), average=False, name="after")
        assert g[0].item() == 2.0
    """, timeout=240)


def test_join_after_cached_allgather_np2():
    """Round-2 regression (segfault): a CACHED allgather slot firing while
    ranks are joined must substitute the sizes the cached response carries
    (reference AllocateZeros(tensor_sizes[i])), not zero rows."""
    run_workers(2, """
        # negotiate + cache an allgather with real contributions
        for i in range(3):
            g = hvd.allgather(torch.full((2, 3), float(rank)), name="cag")
            assert g.shape == (4, 3)
        # now both ranks join; the cached slot may fire with stale sizes
        hvd.join()
        out = hvd.allreduce(torch.ones(4), average=False, name="after")
        assert out[0].item() == 2.0
    """, timeout=240)


def test_cache_capacity_pressure_np3():
    """LRU eviction churn: 40 rotating tensor names against an 8-slot cache
    — slot allocation/eviction must stay bit-identical across ranks (the
    fast path serializes the collective order) while values stay right."""
    run_workers(3, """
        for round_ in range(6):
            for i in range(40):
                t = torch.full((16,), float(rank + 1 + i))
                out = hvd.allreduce(t, average=False, name=f"cap{i}")
                assert out[0].item() == float(3 * (1 + i) + 3), (round_, i)
    """, extra_env={"HOROVOD_CACHE_CAPACITY": "8"}, timeout=300)


def test_many_tensor_flood_np2():
    """300 concurrent async allreduces of mixed sizes: kernel-batch
    splitting (48/launch), fusion-threshold splitting and handle
    bookkeeping under load."""
    run_workers(2, """
        hs = []
        for i in range(300):
            t = torch.full((1 + (i % 97),), float(i + rank))
            hs.append(hvd.allreduce_async(t, average=False, name=f"fl{i}"))
        for i, h in enumerate(hs):
            out = hvd.synchronize(h)
            assert out[0].item() == float(2 * i + 1), (i, out[0].item())
    """, timeout=300)


def test_grouped_under_cache_pressure_np2():
    """Grouped responses + LRU eviction churn: group slots must evict and
    renegotiate coherently (partial-group cache states converge)."""
    run_workers(2, """
        for round_ in range(5):
            for g in range(4):
                ts = [torch.full((8,), float(rank + 1 + g)) for _ in range(3)]
                outs = hvd.grouped_allreduce(ts, average=False,
                                             name=f"grp{g}")
                for o in outs:
                    assert o[0].item() == float(2 * (1 + g) + 1), (round_, g)
            for i in range(10):
                out = hvd.allreduce(torch.ones(4), average=False,
                                    name=f"s{i}")
                assert out[0].item() == 2.0
    """, extra_env={"HOROVOD_CACHE_CAPACITY": "8"}, timeout=240)


def test_signature_churn_np2():
    """Signature-invalidation churn: rotating alltoall splits, rotating
    broadcast roots and alternating dtypes under one name must renegotiate
    each time (INVALID path) without desync."""
    run_workers(2, """
        for i in range(10):
            a, b = (1 + i % 3), (3 - i % 3)
            t = torch.arange((a + b) * 2, dtype=torch.float32).reshape(-1, 2)
            out, rs = hvd.alltoall(t, splits=torch.tensor([a, b]),
                                   name="rot")
            assert out.shape[0] == (2 * a if rank == 0 else 2 * b), i
        for i in range(6):
            root = i % 2
            o = hvd.broadcast(torch.full((4,), float(rank + 1)),
                              root_rank=root, name="rootrot")
            assert o[0].item() == float(root + 1), i
        for i in range(6):
            dt = torch.float32 if i % 2 == 0 else torch.float64
            o = hvd.allreduce(torch.ones(3, dtype=dt), average=False,
                              name="dtrot")
            assert o[0].item() == 2.0 and o.dtype == dt, i
    """, timeout=240)


def test_mixed_barrier_subset_interleaving_np4():
    """Randomized (rank-identical) interleaving of barriers, global
    allreduces and three overlapping subset allreduces at np=4."""
    run_workers(4, """
        import random
        rng = random.Random(42)
        sets = [hvd.add_process_set(hvd.ProcessSet([0, 1])),
                hvd.add_process_set(hvd.ProcessSet([2, 3])),
                hvd.add_process_set(hvd.ProcessSet([0, 2, 3]))]
        members = [[0, 1], [2, 3], [0, 2, 3]]
        for i in range(30):
            k = rng.randint(0, 3)
            if k == 0:
                hvd.barrier()
            elif k == 1:
                out = hvd.allreduce(torch.ones(32) * (rank + 1),
                                    average=False, name=f"g{i}")
                assert out[0].item() == 10.0
            else:
                si = rng.randint(0, 2)
                if rank in members[si]:
                    out = hvd.allreduce(torch.ones(8), average=False,
                                        name=f"s{si}.{i}",
                                        process_set=sets[si])
                    assert out[0].item() == float(len(members[si]))
        hvd.barrier()
    """, timeout=240)
