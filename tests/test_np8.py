"""np=8 CPU fuzz — pre-hardening for the driver's 8-GPU runs.

Cross-checks every collective at the full node width (8 ranks) against
torch.distributed/gloo or a local float64 simulation, with randomized
shapes/splits, exercising exactly the spots the round-1 verdict flagged as
risky at n>1: allgather-v offsets at li>0, alltoall split-matrix indexing,
reducescatter-v remainders, and the VHDD combine order at np=8.
"""
import pytest
import torch

from tests.parallel_util import run_workers


def test_np8_fuzz_vs_torch_distributed():
    from horovod_amd.runner.launch import find_free_port
    port = find_free_port()
    run_workers(8, f"""
        import os
        import torch.distributed as dist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ["MASTER_PORT"] = "{port}"
        dist.init_process_group("gloo", rank=rank, world_size=size)
        g = torch.Generator().manual_seed(42)  # SHARED stream: same draws
        gr = torch.Generator().manual_seed(1000 + rank)  # per-rank data
        for trial in range(3):
            cols = int(torch.randint(1, 5, (1,), generator=g))
            # --- allreduce (uneven tensor count per cycle stresses fusion)
            nt = int(torch.randint(1, 4, (1,), generator=g))
            ts = [torch.randn(int(torch.randint(1, 2000, (1,), generator=g)),
                              cols, generator=gr)
                  for _ in range(nt)]
            ours = [hvd.allreduce_async(t, average=False,
                                        name=f"f.ar.{{trial}}.{{i}}")
                    for i, t in enumerate(ts)]
            for t, h in zip(ts, ours):
                o = hvd.synchronize(h)
                ref = t.clone(); dist.all_reduce(ref)
                assert torch.allclose(o, ref, atol=1e-5), "allreduce"
            # --- allgather-v: random rows per rank (offset math at li>0)
            rows = [int(torch.randint(0, 6, (1,), generator=g))
                    for _ in range(size)]
            t = torch.randn(rows[rank], cols, generator=gr)
            o = hvd.allgather(t, name=f"f.ag.{{trial}}")
            parts = [torch.empty(0) for _ in range(size)]
            # gloo all_gather needs same shapes; oracle via object gather
            objs = [None] * size
            dist.all_gather_object(objs, t)
            ref = torch.cat([x for x in objs if x.numel() or True])
            assert o.shape == ref.shape and torch.allclose(o, ref,
                                                           atol=1e-6), "agv"
            # --- alltoall-v: random split matrix row (mine)
            splits = [int(torch.randint(0, 4, (1,), generator=gr))
                      for _ in range(size)]
            t = torch.randn(sum(splits), cols, generator=gr)
            o, rs = hvd.alltoall(t, splits=torch.tensor(splits),
                                 name=f"f.a2a.{{trial}}")
            allsplits = [None] * size
            dist.all_gather_object(allsplits, (splits, t))
            exp_chunks = []
            for r in range(size):
                sp, data = allsplits[r]
                off = sum(sp[:rank])
                exp_chunks.append(data[off:off + sp[rank]])
            exp = torch.cat(exp_chunks) if exp_chunks else t[:0]
            assert rs.tolist() == [allsplits[r][0][rank]
                                   for r in range(size)], "a2a splits"
            assert o.shape == exp.shape and torch.allclose(o, exp,
                                                           atol=1e-6), "a2a"
            # --- reducescatter-v: first dim with remainder
            first = size * 2 + int(torch.randint(0, size, (1,), generator=g))
            t = torch.randn(first, cols, generator=gr)
            o = hvd.reducescatter(t, op=hvd.Sum, name=f"f.rs.{{trial}}")
            full = t.clone(); dist.all_reduce(full)
            base, rem = divmod(first, size)
            my_rows = base + (1 if rank < rem else 0)
            row_off = rank * base + min(rank, rem)
            assert torch.allclose(o, full[row_off:row_off + my_rows],
                                  atol=1e-5), "rsv"
            # --- broadcast from a rotating root
            root = trial % size
            t = torch.randn(5, cols, generator=gr)
            o = hvd.broadcast(t, root_rank=root, name=f"f.bc.{{trial}}")
            ref = t.clone(); dist.broadcast(ref, src=root)
            assert torch.allclose(o, ref, atol=1e-6), "bcast"
        dist.destroy_process_group()
    """, timeout=600)


@pytest.mark.parametrize("np_", [5, 8])
def test_np_adasum_golden(np_):
    """VHDD combine order at non-power-of-2 (np=5) and full width (np=8)
    against a float64 simulation of core.cc's tree."""
    run_workers(np_, """
        n = size
        def vec(r):
            i = torch.arange(123, dtype=torch.float64)
            return torch.cos(i * (r + 2)) + 0.05 * r
        def combine(a, b):
            dot = (a * b).sum(); na = (a * a).sum(); nb = (b * b).sum()
            ac = 1.0 - dot / (2 * na) if na > 0 else 1.0
            bc = 1.0 - dot / (2 * nb) if nb > 0 else 1.0
            return a * ac + b * bc
        work = [vec(r) for r in range(n)]
        p = 1
        while p * 2 <= n:
            p *= 2
        for i in range(p, n):
            work[i - p] = combine(work[i - p], work[i])
        stride = 1
        while stride < p:
            for i in range(0, p - stride, 2 * stride):
                work[i] = combine(work[i], work[i + stride])
            stride *= 2
        expected = work[0].float()
        out = hvd.allreduce(vec(rank).float(), op=hvd.Adasum, name="ada8")
        assert torch.allclose(out, expected, rtol=1e-5, atol=1e-6), \
            (out - expected).abs().max()
    """, timeout=420)


def test_np8_subsets_and_grouped():
    """Concurrent subset collectives + grouped ops at np=8: response-stream
    determinism under three overlapping process sets."""
    run_workers(8, """
        evens = hvd.add_process_set(hvd.ProcessSet([0, 2, 4, 6]))
        odds = hvd.add_process_set(hvd.ProcessSet([1, 3, 5, 7]))
        half = hvd.add_process_set(hvd.ProcessSet([0, 1, 2, 3]))
        mine = evens if rank % 2 == 0 else odds
        my_members = [0, 2, 4, 6] if rank % 2 == 0 else [1, 3, 5, 7]
        for trial in range(4):
            t = torch.full((64,), float(rank + 1))
            h1 = hvd.allreduce_async(t, average=False,
                                     name=f"s.a.{trial}", process_set=mine)
            h2 = None
            if rank < 4:
                ts = [torch.full((16,), float(rank + 1)) for _ in range(3)]
                h2 = hvd.grouped_allreduce_async(ts, average=False,
                                                 name=f"s.g.{trial}",
                                                 process_set=half)
            o1 = hvd.synchronize(h1)
            exp = float(sum(r + 1 for r in my_members))
            assert torch.allclose(o1, torch.full_like(o1, exp)), o1[0]
            if h2 is not None:
                outs = [hvd.synchronize(h) for h in h2] if isinstance(
                    h2, list) else hvd.synchronize(h2)
                for o in outs:
                    assert torch.allclose(o, torch.full_like(o, 10.0)), o[0]
    """, timeout=420)
