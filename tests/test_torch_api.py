"""Framework-level tests: DistributedOptimizer training convergence across
ranks, state broadcast helpers, SyncBatchNorm numerics, compression,
autograd of collectives, Adasum golden numerics.
(Reference coverage model: test/parallel/test_torch.py + test_adasum_pytorch.py.)
"""
import numpy as np
import pytest
import torch

from tests.parallel_util import run_workers


def test_distributed_optimizer_sync_np2():
    """After N steps on different data, parameters must be identical across
    ranks and follow the average-gradient trajectory."""
    run_workers(2, """
        torch.manual_seed(1234)           # same init on all ranks
        model = torch.nn.Sequential(
            torch.nn.Linear(10, 16), torch.nn.ReLU(), torch.nn.Linear(16, 1))
        opt = torch.optim.SGD(model.parameters(), lr=0.05)
        opt = hvd.DistributedOptimizer(
            opt, named_parameters=model.named_parameters())
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(100 + rank)     # different data per rank
        for step in range(5):
            x = torch.randn(8, 10)
            y = torch.randn(8, 1)
            opt.zero_grad()
            loss = ((model(x) - y) ** 2).mean()
            loss.backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        gathered = hvd.allgather(flat.unsqueeze(0), name="final_params")
        assert torch.allclose(gathered[0], gathered[1], atol=1e-6), \
            (gathered[0] - gathered[1]).abs().max()
    """)


def test_distributed_optimizer_groups_np2():
    run_workers(2, """
        torch.manual_seed(7)
        model = torch.nn.Sequential(
            torch.nn.Linear(6, 8), torch.nn.ReLU(), torch.nn.Linear(8, 2))
        opt = hvd.DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.1),
            named_parameters=model.named_parameters(), groups=2)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(100 + rank)
        for step in range(3):
            opt.zero_grad()
            loss = model(torch.randn(4, 6)).sum()
            loss.backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        g = hvd.allgather(flat.unsqueeze(0), name="gp")
        assert torch.allclose(g[0], g[1], atol=1e-6)
    """)


def test_backward_passes_per_step_np2():
    run_workers(2, """
        torch.manual_seed(3)
        model = torch.nn.Linear(4, 1)
        opt = hvd.DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.1),
            named_parameters=model.named_parameters(),
            backward_passes_per_step=2)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(50 + rank)
        for step in range(2):
            opt.zero_grad()
            for micro in range(2):
                loss = model(torch.randn(4, 4)).sum()
                loss.backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        g = hvd.allgather(flat.unsqueeze(0), name="bpps")
        assert torch.allclose(g[0], g[1], atol=1e-6)
    """)


def test_compression_fp16_np2():
    run_workers(2, """
        from horovod_amd.torch.compression import Compression
        t = torch.randn(1000) * (rank + 1)
        out = hvd.allreduce(t, average=False, compression=Compression.fp16,
                            name="comp")
        ref = hvd.allreduce(t, average=False, name="nocomp")
        assert torch.allclose(out, ref, rtol=1e-2, atol=1e-2), \
            (out - ref).abs().max()
    """)


def test_broadcast_object_np2():
    run_workers(2, """
        from horovod_amd.torch import broadcast_object, allgather_object
        obj = {"a": 1, "b": [1, 2, 3]} if rank == 0 else None
        got = broadcast_object(obj, root_rank=0)
        assert got == {"a": 1, "b": [1, 2, 3]}, got
        objs = allgather_object({"rank": rank})
        assert objs == [{"rank": 0}, {"rank": 1}], objs
    """)


def test_broadcast_optimizer_state_np2():
    run_workers(2, """
        torch.manual_seed(10 + rank)  # deliberately different init
        model = torch.nn.Linear(5, 3)
        opt = torch.optim.Adam(model.parameters(), lr=0.01)
        # build some state
        loss = model(torch.randn(2, 5)).sum()
        loss.backward()
        opt.step()
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        from horovod_amd.torch import broadcast_optimizer_state
        broadcast_optimizer_state(opt, root_rank=0)
        # verify state equality via allgather of exp_avg
        s = opt.state[list(model.parameters())[0]]["exp_avg"].flatten()
        g = hvd.allgather(s.unsqueeze(0), name="st")
        assert torch.allclose(g[0], g[1], atol=1e-7)
    """)


def test_sync_batch_norm_np2():
    """SyncBatchNorm over 2 ranks must equal BatchNorm over the concatenated
    batch (reference: sync_batch_norm tests)."""
    run_workers(2, """
        torch.manual_seed(5)
        full = torch.randn(8, 3, 4, 4)
        mine = full[rank * 4:(rank + 1) * 4].clone().requires_grad_(True)

        sbn = hvd.SyncBatchNorm(3, momentum=0.3)
        bn = torch.nn.BatchNorm2d(3, momentum=0.3)
        out = sbn(mine)
        ref_in = full.clone().requires_grad_(True)
        ref = bn(ref_in)
        assert torch.allclose(out, ref[rank * 4:(rank + 1) * 4], atol=1e-5), \
            (out - ref[rank * 4:(rank + 1) * 4]).abs().max()
        assert torch.allclose(sbn.running_mean, bn.running_mean, atol=1e-5)
        assert torch.allclose(sbn.running_var, bn.running_var, atol=1e-4)
        # backward equivalence
        gout = torch.randn_like(full)
        out.backward(gout[rank * 4:(rank + 1) * 4])
        ref.backward(gout)
        assert torch.allclose(mine.grad,
                              ref_in.grad[rank * 4:(rank + 1) * 4],
                              atol=1e-4), \
            (mine.grad - ref_in.grad[rank*4:(rank+1)*4]).abs().max()
    """)


def _adasum_golden(tensors):
    """NumPy golden model of the VHDD combine tree (matches core.cc and
    adasum_kernels.hip; reference numerics: adasum.h:396-412)."""
    def combine(a, b):
        dot = float(np.dot(a, b))
        na = float(np.dot(a, a))
        nb = float(np.dot(b, b))
        ac = 1.0 - dot / (2 * na) if na > 0 else 1.0
        bc = 1.0 - dot / (2 * nb) if nb > 0 else 1.0
        return ac * a + bc * b

    work = [t.astype(np.float64) for t in tensors]
    p = 1
    while p * 2 <= len(work):
        p *= 2
    for i in range(p, len(work)):
        work[i - p] = combine(work[i - p], work[i])
    work = work[:p]
    stride = 1
    while stride < p:
        i = 0
        while i + stride < p:
            work[i] = combine(work[i], work[i + stride])
            i += 2 * stride
        stride *= 2
    return work[0]


def test_adasum_golden_np2():
    rng = np.random.RandomState(42)
    a = rng.randn(64).astype(np.float32)
    b = rng.randn(64).astype(np.float32)
    expected = _adasum_golden([a, b])
    run_workers(2, f"""
        import numpy as np
        vecs = [np.array({a.tolist()!r}, dtype=np.float32),
                np.array({b.tolist()!r}, dtype=np.float32)]
        t = torch.from_numpy(vecs[rank]).clone()
        out = hvd.allreduce(t, op=hvd.Adasum, name="adasum")
        expected = np.array({expected.tolist()!r})
        assert np.allclose(out.numpy(), expected, atol=1e-5), \\
            np.abs(out.numpy() - expected).max()
    """)


def test_adasum_golden_np3():
    rng = np.random.RandomState(7)
    vecs = [rng.randn(33).astype(np.float32) for _ in range(3)]
    expected = _adasum_golden(vecs)
    run_workers(3, f"""
        import numpy as np
        allv = {[v.tolist() for v in vecs]!r}
        t = torch.tensor(allv[rank], dtype=torch.float32)
        out = hvd.allreduce(t, op=hvd.Adasum, name="adasum3")
        expected = np.array({expected.tolist()!r})
        assert np.allclose(out.numpy(), expected, atol=1e-5), \\
            np.abs(out.numpy() - expected).max()
    """)


def test_autograd_allgather_np2():
    run_workers(2, """
        x = (torch.arange(4).float() + rank).requires_grad_(True)
        y = hvd.allgather(x, name="ag_grad")
        loss = (y * torch.arange(8).float()).sum()
        loss.backward()
        expected = torch.arange(rank * 4, rank * 4 + 4).float() * 2
        assert torch.allclose(x.grad, expected), (x.grad, expected)
    """)


def test_autograd_broadcast_np2():
    run_workers(2, """
        x = torch.ones(3, requires_grad=True)
        y = hvd.broadcast(x, root_rank=0, name="bc_grad")
        y.sum().backward()
        if rank == 0:
            assert torch.allclose(x.grad, torch.full((3,), 2.0)), x.grad
        else:
            assert torch.allclose(x.grad, torch.zeros(3)), x.grad
    """)


def test_sparse_allreduce_np2():
    run_workers(2, """
        i = torch.tensor([[0, 2]]) if rank == 0 else torch.tensor([[1, 2]])
        v = torch.tensor([1.0, 2.0]) if rank == 0 else torch.tensor([3.0, 4.0])
        sp = torch.sparse_coo_tensor(i, v, (4,))
        closure = hvd.sparse_allreduce_async(sp, name="sp", op=hvd.Sum)
        out = closure().to_dense()
        assert torch.allclose(out, torch.tensor([1.0, 3.0, 6.0, 0.0])), out
    """)


def test_timeline_np2(tmp_path):
    """Runtime timeline start/stop produces valid Chrome-trace JSON on every
    rank (reference: test_timeline.py)."""
    import json as _json
    tl = str(tmp_path / "tl.json")
    run_workers(2, f"""
        hvd.start_timeline({tl!r})
        for i in range(3):
            hvd.allreduce(torch.ones(100), average=False, name=f"tl{{i}}")
        hvd.stop_timeline()
    """)
    data = _json.load(open(tl))
    names = [e.get("name") for e in data]
    assert "ALLREDUCE" in names, names[:10]
    assert "NEGOTIATE" in names, "per-tensor negotiate phases missing"
    data1 = _json.load(open(tl + ".1"))
    assert data1


def test_autotune_np2(tmp_path):
    log = str(tmp_path / "autotune.csv")
    run_workers(2, """
        import horovod_amd._core as core
        before = core.get_fusion_threshold()
        for i in range(200):
            hvd.allreduce(torch.ones(4096), average=False, name="at")
        # autotuner window is 3s; this loop is fast, so just assert the
        # mechanism is alive: threshold is a sane positive value
        assert core.get_fusion_threshold() > 0
    """, extra_env={"HOROVOD_AUTOTUNE": "1", "HOROVOD_AUTOTUNE_LOG": log})


def test_adasum_optimizer_np2():
    """op=Adasum routes to the delta optimizer; params stay identical across
    ranks (reference: _DistributedAdasumOptimizer)."""
    run_workers(2, """
        torch.manual_seed(21)
        model = torch.nn.Linear(6, 2)
        opt = hvd.DistributedOptimizer(
            torch.optim.Adam(model.parameters(), lr=0.01),
            named_parameters=model.named_parameters(), op=hvd.Adasum)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(77 + rank)
        for step in range(4):
            opt.zero_grad()
            loss = model(torch.randn(5, 6)).sum()
            loss.backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        g = hvd.allgather(flat.unsqueeze(0), name="ada_opt")
        assert torch.allclose(g[0], g[1], atol=1e-6), \
            (g[0] - g[1]).abs().max()
    """)


def test_optimizer_with_process_set_np3():
    """DistributedOptimizer over a 2-rank subset: members train in lockstep,
    the outsider trains independently."""
    run_workers(3, """
        ps = hvd.add_process_set(hvd.ProcessSet([0, 1]))
        torch.manual_seed(4)
        model = torch.nn.Linear(5, 1)
        if rank in (0, 1):
            opt = hvd.DistributedOptimizer(
                torch.optim.SGD(model.parameters(), lr=0.1),
                named_parameters=model.named_parameters(), process_set=ps)
            hvd.broadcast_parameters(model.state_dict(), root_rank=0,
                                     process_set=ps)
            torch.manual_seed(10 + rank)
            for _ in range(3):
                opt.zero_grad()
                model(torch.randn(4, 5)).sum().backward()
                opt.step()
            flat = torch.cat([p.detach().flatten()
                              for p in model.parameters()])
            g = hvd.allgather(flat.unsqueeze(0), process_set=ps, name="sub_p")
            assert torch.allclose(g[0], g[1], atol=1e-6)
        hvd.barrier()
    """)


def test_mismatched_dtype_and_op_errors_np2():
    run_workers(2, """
        t = torch.ones(4) if rank == 0 else torch.ones(4, dtype=torch.float64)
        try:
            hvd.allreduce(t, average=False, name="dt_bad")
            raise SystemExit("expected dtype mismatch error")
        except RuntimeError as e:
            assert "data types" in str(e) or "Mismatched" in str(e), e
        op = hvd.Sum if rank == 0 else hvd.Max
        try:
            hvd.allreduce(torch.ones(4), op=op, name="op_bad")
            raise SystemExit("expected op mismatch error")
        except RuntimeError as e:
            assert "Mismatched" in str(e), e
    """)


def test_gradient_predivide_np2():
    run_workers(2, """
        torch.manual_seed(2)
        model = torch.nn.Linear(3, 1)
        opt = hvd.DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.1),
            named_parameters=model.named_parameters(),
            gradient_predivide_factor=2.0)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        ref = [p.detach().clone() for p in model.parameters()]
        x = torch.ones(2, 3)
        opt.zero_grad()
        model(x).sum().backward()
        # expected grad: averaged over 2 identical ranks == local grad
        expected = [p.grad.clone() for p in model.parameters()]
        opt.step()
        for p, r, e in zip(model.parameters(), ref, expected):
            assert torch.allclose(p.detach(), r - 0.1 * e, atol=1e-6)
    """)


def test_autotune_propagates_np2(tmp_path):
    """With a short sampling window, the coordinator must propose and
    propagate new fusion parameters via TUNE responses."""
    log = str(tmp_path / "at.csv")
    run_workers(2, """
        import time
        import horovod_amd._core as core
        start = (core.get_fusion_threshold(), core.get_cycle_time_ms())
        changed = False
        for i in range(4000):
            hvd.allreduce(torch.ones(8192), average=False, name="atp")
            if (core.get_fusion_threshold(),
                    core.get_cycle_time_ms()) != start:
                changed = True
                break
        assert changed, "no TUNE response arrived"
    """, extra_env={"HOROVOD_AUTOTUNE": "1",
                    "HOROVOD_AUTOTUNE_WINDOW_SECONDS": "0.3",
                    "HOROVOD_AUTOTUNE_LOG": log}, timeout=300)
    assert "," in open(log).read()


def test_fuzz_collective_sequences_np2():
    """Randomized (rank-identical) sequences of mixed collectives; results
    checked against locally computable expectations."""
    run_workers(2, """
        import random
        rng = random.Random(1234)        # identical schedule on both ranks
        for i in range(60):
            kind = rng.choice(["allreduce", "allgather", "broadcast",
                               "reducescatter", "barrier", "grouped"])
            n = rng.randint(1, 300)
            base = torch.arange(n).float()
            mine = base * (rank + 1)
            if kind == "allreduce":
                out = hvd.allreduce(mine, average=False, name=f"f{i}")
                assert torch.allclose(out, base * 3), (i, kind)
            elif kind == "allgather":
                out = hvd.allgather(mine, name=f"f{i}")
                assert torch.allclose(out, torch.cat([base, base * 2])), i
            elif kind == "broadcast":
                root = rng.randint(0, 1)
                out = hvd.broadcast(mine, root_rank=root, name=f"f{i}")
                assert torch.allclose(out, base * (root + 1)), i
            elif kind == "reducescatter":
                out = hvd.reducescatter(mine, op=hvd.Sum, name=f"f{i}")
                lo = 0 if rank == 0 else (n + 1) // 2
                hi = (n + 1) // 2 if rank == 0 else n
                assert torch.allclose(out, base[lo:hi] * 3), i
            elif kind == "barrier":
                hvd.barrier()
            else:
                outs = hvd.grouped_allreduce([mine, mine + 1], average=False,
                                             name=f"f{i}")
                assert torch.allclose(outs[1], base * 3 + 2), i
    """, timeout=300)


def test_multithreaded_enqueue_np2():
    """Two python threads enqueue DISTINCT named ops concurrently; the queue
    and handle manager must be thread-safe.  (Cross-rank op ORDER still must
    match, so each thread owns its own name space.)"""
    run_workers(2, """
        import threading
        errors = []
        def worker(tag):
            try:
                for i in range(30):
                    out = hvd.allreduce(torch.ones(64), average=False,
                                        name=f"mt.{tag}.{i}")
                    assert out.sum().item() == 128.0
            except Exception as e:
                errors.append(e)
        ts = [threading.Thread(target=worker, args=(t,)) for t in "ab"]
        for t in ts: t.start()
        for t in ts: t.join()
        assert not errors, errors
    """, timeout=300)


def test_init_with_process_sets_np2():
    run_workers(2, """
        # (re-init path: hvd.init already ran in the harness prelude, so
        # register via the documented collective API instead)
        ps = hvd.add_process_set([0, 1])
        out = hvd.allreduce(torch.ones(2), average=False, process_set=ps,
                            name="ip")
        assert out.sum().item() == 4.0
    """)


def test_metric_averaging_np2():
    run_workers(2, """
        from horovod_amd.torch import MetricAverager, avg_metrics
        out = avg_metrics({"loss": 1.0 + rank, "acc": 0.5 * (rank + 1)})
        assert abs(out["loss"] - 1.5) < 1e-9 and abs(out["acc"] - 0.75) < 1e-9
        m = MetricAverager()
        m.update("loss", 2.0 + rank, n=rank + 1)  # weighted
        avg = m.averages()
        # (2.0*1 + 3.0*2) / 3 = 8/3
        assert abs(avg["loss"] - 8.0 / 3) < 1e-9, avg
    """)


def test_warmup_scheduler_np2():
    run_workers(2, """
        from horovod_amd.torch import WarmupScheduler
        model = torch.nn.Linear(2, 1)
        opt = torch.optim.SGD(model.parameters(), lr=0.8)
        sched = WarmupScheduler(opt, warmup_steps=4)
        lrs = []
        for _ in range(6):
            lrs.append(opt.param_groups[0]["lr"])
            opt.step()
            sched.step()
        # starts at lr/size = 0.4, ramps to 0.8, stays
        assert abs(lrs[0] - 0.4) < 1e-9, lrs
        assert abs(lrs[-1] - 0.8) < 1e-9, lrs
        assert lrs == sorted(lrs), lrs
    """)


def test_adasum_golden_np4():
    rng = np.random.RandomState(11)
    vecs = [rng.randn(48).astype(np.float32) for _ in range(4)]
    expected = _adasum_golden(vecs)
    run_workers(4, f"""
        import numpy as np
        allv = {[v.tolist() for v in vecs]!r}
        t = torch.tensor(allv[rank], dtype=torch.float32)
        out = hvd.allreduce(t, op=hvd.Adasum, name="adasum4")
        expected = np.array({expected.tolist()!r})
        assert np.allclose(out.numpy(), expected, atol=1e-5), \\
            np.abs(out.numpy() - expected).max()
    """)


def test_optimizer_groups_with_compression_np2():
    run_workers(2, """
        from horovod_amd.torch.compression import Compression
        torch.manual_seed(13)
        model = torch.nn.Sequential(torch.nn.Linear(8, 16),
                                    torch.nn.ReLU(), torch.nn.Linear(16, 4))
        opt = hvd.DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.05),
            named_parameters=model.named_parameters(), groups=2,
            compression=Compression.fp16)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(40 + rank)
        for _ in range(3):
            opt.zero_grad()
            model(torch.randn(4, 8)).sum().backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        g = hvd.allgather(flat.unsqueeze(0), name="gc")
        assert torch.allclose(g[0], g[1], atol=1e-5), \
            (g[0] - g[1]).abs().max()
    """)


def test_adasum_golden_np5():
    rng = np.random.RandomState(3)
    vecs = [rng.randn(16).astype(np.float32) for _ in range(5)]
    expected = _adasum_golden(vecs)
    run_workers(5, f"""
        import numpy as np
        allv = {[v.tolist() for v in vecs]!r}
        t = torch.tensor(allv[rank], dtype=torch.float32)
        out = hvd.allreduce(t, op=hvd.Adasum, name="adasum5")
        expected = np.array({expected.tolist()!r})
        assert np.allclose(out.numpy(), expected, atol=1e-5)
    """)


def test_collectives_np5():
    run_workers(5, """
        out = hvd.allreduce(torch.ones(3) * (rank + 1), average=False,
                            name="n5")
        assert out[0].item() == 15.0
        g = hvd.allgather(torch.full((1,), float(rank)), name="n5g")
        assert g.tolist() == [0.0, 1.0, 2.0, 3.0, 4.0]
        rs = hvd.reducescatter(torch.arange(7).float(), op=hvd.Sum,
                               name="n5rs")
        # 7 rows over 5 ranks: [2,2,1,1,1]
        expected_rows = [2, 2, 1, 1, 1][rank]
        assert rs.numel() == expected_rows, rs
    """)


def test_groups_with_backward_passes_np2():
    run_workers(2, """
        torch.manual_seed(17)
        model = torch.nn.Sequential(torch.nn.Linear(4, 6), torch.nn.ReLU(),
                                    torch.nn.Linear(6, 2))
        opt = hvd.DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.1),
            named_parameters=model.named_parameters(), groups=2,
            backward_passes_per_step=2)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(60 + rank)
        for step in range(2):
            opt.zero_grad()
            for micro in range(2):
                model(torch.randn(3, 4)).sum().backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        g = hvd.allgather(flat.unsqueeze(0), name="gbp")
        assert torch.allclose(g[0], g[1], atol=1e-6), \
            (g[0] - g[1]).abs().max()
    """)


def test_autotune_categorical_np2(tmp_path):
    """Round-2: the autotuner explores the categorical arms (cache on/off +
    one-shot crossover) — log rows carry 5 fields and results stay correct
    while the cache-off arm renegotiates every cycle."""
    log = str(tmp_path / "atc.csv")
    run_workers(2, """
        import time
        deadline = time.time() + 25
        i = 0
        while time.time() < deadline and i < 6000:
            out = hvd.allreduce(torch.full((2048,), float(rank + 1)),
                                average=False, name="atc")
            assert out[0].item() == 3.0, out[0].item()
            i += 1
    """, extra_env={"HOROVOD_AUTOTUNE": "1",
                    "HOROVOD_AUTOTUNE_WINDOW_SECONDS": "0.1",
                    "HOROVOD_AUTOTUNE_LOG": log}, timeout=300)
    rows = [ln for ln in open(log).read().splitlines()
            if ln and not ln.startswith(("CONVERGED", "REOPENED"))]
    assert rows, "autotune log empty"
    assert all(len(r.split(",")) == 5 for r in rows), rows[:3]
    # the warm-start grid pins a cache-off probe; with >= 13 windows it ran
    if len(rows) >= 13:
        assert any(r.split(",")[3] == "0" for r in rows), \
            "cache-off arm never explored"


def test_grouped_allreduce_autograd_np2():
    """Round-2: grouped allreduce is differentiable (reference
    HorovodGroupedAllreduce) — one fused negotiation unit whose gradient
    is the grouped allreduce of the incoming grads."""
    run_workers(2, """
        ts = [torch.full((8,), float(rank + 1), requires_grad=True)
              for _ in range(3)]
        outs = hvd.grouped_allreduce(ts, average=False, name="gar")
        for o in outs:
            assert torch.allclose(o, torch.full((8,), 3.0))
        loss = sum((o * (i + 1)).sum() for i, o in enumerate(outs))
        loss.backward()
        # d(loss)/d(t_i) = allreduce-sum of (i+1)*ones = 2*(i+1)
        for i, t in enumerate(ts):
            assert torch.allclose(t.grad, torch.full((8,), 2.0 * (i + 1))), \
                (i, t.grad)
        # grouped allgather differentiable path
        gs = [torch.full((2, 2), float(rank), requires_grad=True)
              for _ in range(2)]
        outs = hvd.grouped_allgather(gs, name="gag")
        sum(o.sum() for o in outs).backward()
        for g in gs:
            assert g.grad is not None and torch.allclose(
                g.grad, torch.full((2, 2), 2.0))
    """)


def test_api_surface_shims_np2():
    run_workers(2, """
        assert hvd.process_set_included(0) == 1
        ps = hvd.add_process_set(hvd.ProcessSet([0]))
        assert hvd.process_set_included(ps.process_set_id) == (
            1 if rank == 0 else 0)
        assert hvd.handle_average_backwards_compatibility(None, True) == \
            hvd.Average
        assert hvd.handle_average_backwards_compatibility(None, False) == \
            hvd.Sum
        assert hvd.handle_average_backwards_compatibility(hvd.Max, None) == \
            hvd.Max
    """)


def test_kitchen_sink_training_np3():
    """Dogfood: sparse embedding grads + dense conv/linear params with
    gradient groups, backward_passes_per_step=2, bf16 wire compression,
    metric averaging over a subset, and a mid-run broadcast — all in one
    3-rank training loop that must keep ranks bit-identical."""
    run_workers(3, """
        torch.manual_seed(3)
        emb = torch.nn.Embedding(50, 8, sparse=True)
        mlp = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                  torch.nn.Linear(16, 4))
        params = list(emb.parameters()) + list(mlp.parameters())
        names = list(emb.named_parameters()) + list(mlp.named_parameters())
        opt = torch.optim.SGD(params, lr=0.05)
        opt = hvd.DistributedOptimizer(
            opt, named_parameters=names, backward_passes_per_step=2,
            compression=hvd.Compression.bf16, groups=2)
        hvd.broadcast_parameters(emb.state_dict(), root_rank=0,
                                 prefix="emb")
        hvd.broadcast_parameters(mlp.state_dict(), root_rank=0,
                                 prefix="mlp")
        ps = hvd.add_process_set(hvd.ProcessSet([0, 2]))
        from horovod_amd.torch.metrics import avg_metrics
        for step in range(6):
            for micro in range(2):
                idx = torch.randint(0, 50, (16,),
                                    generator=torch.Generator()
                                    .manual_seed(step * 10 + micro))
                target = torch.randn(16, 4,
                                     generator=torch.Generator()
                                     .manual_seed(99 + step))
                out = mlp(emb(idx))
                loss = torch.nn.functional.mse_loss(out, target)
                if micro == 0:
                    opt.zero_grad()
                loss.backward()
            opt.step()
            if rank in (0, 2):
                m = avg_metrics({"loss": float(loss)}, process_set=ps)
                assert m["loss"] > 0
        # every rank must hold identical weights (incl sparse embedding)
        flat = torch.cat([p.detach().flatten() for p in params])
        mx = hvd.allreduce(flat, op=hvd.Max, name="ksmax")
        mn = hvd.allreduce(flat, op=hvd.Min, name="ksmin")
        assert torch.allclose(mx, mn, atol=1e-6), \
            (mx - mn).abs().max().item()
    """, timeout=300)


def test_sbn_model_equivalence_np2():
    """Gold-standard dogfood: a conv+SyncBatchNorm model trained at np=2
    (per-rank batch B, gradient averaging) must produce the same weights
    as the identical model trained single-process on the concatenated
    batch 2B with plain BatchNorm — step for step."""
    run_workers(2, """
        import torch.nn as nn
        from horovod_amd.torch.sync_batch_norm import SyncBatchNorm
        torch.manual_seed(11)
        torch.use_deterministic_algorithms(True)

        def make(bn):
            torch.manual_seed(17)
            return nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), bn(8),
                                 nn.ReLU(), nn.Flatten(),
                                 nn.Linear(8 * 8 * 8, 5))

        dist = make(SyncBatchNorm)
        ref = make(nn.BatchNorm2d)
        for pd, pr in zip(dist.parameters(), ref.parameters()):
            assert torch.equal(pd, pr)
        opt_d = hvd.DistributedOptimizer(
            torch.optim.SGD(dist.parameters(), lr=0.1),
            named_parameters=dist.named_parameters())
        opt_r = torch.optim.SGD(ref.parameters(), lr=0.1)

        for step in range(4):
            g = torch.Generator().manual_seed(200 + step)
            full_x = torch.randn(8, 3, 8, 8, generator=g)
            full_t = torch.randint(0, 5, (8,), generator=g)
            my_x = full_x[rank * 4:(rank + 1) * 4]
            my_t = full_t[rank * 4:(rank + 1) * 4]
            opt_d.zero_grad()
            torch.nn.functional.cross_entropy(dist(my_x), my_t).backward()
            opt_d.step()
            # oracle: one process, the whole batch
            opt_r.zero_grad()
            torch.nn.functional.cross_entropy(ref(full_x), full_t).backward()
            opt_r.step()
        for (n, pd), pr in zip(dist.named_parameters(), ref.parameters()):
            assert torch.allclose(pd, pr, rtol=1e-4, atol=1e-5), \
                (n, (pd - pr).abs().max().item())
        # running stats must match the full-batch oracle too
        assert torch.allclose(dist[1].running_mean, ref[1].running_mean,
                              atol=1e-5)
        assert torch.allclose(dist[1].running_var, ref[1].running_var,
                              atol=1e-4)
    """, timeout=300)


def test_autograd_alltoall_reducescatter_np2():
    """Gradient definitions: alltoall grad = reverse alltoall;
    reducescatter grad = allgather of the incoming shard grads."""
    run_workers(2, """
        # alltoall: rank r sends rows valued 10r+dest
        t = torch.stack([torch.full((2,), 10.0 * rank + d)
                         for d in (0, 0, 1)]).requires_grad_(True)
        splits = torch.tensor([2, 1]) if rank == 0 else torch.tensor([2, 1])
        out, rs = hvd.alltoall(t, splits=splits, name="a2ag")
        # weight received rows by (rank+1); backward reverses the exchange
        (out.sum() * (rank + 1)).backward()
        # my row j went to dest d(j); grad = (d(j)+1)
        exp = torch.tensor([[1.0, 1.0], [1.0, 1.0], [2.0, 2.0]])
        assert torch.allclose(t.grad, exp), t.grad
        # reducescatter: grad of my shard allgathers back
        x = torch.ones(4, 3, requires_grad=True)
        y = hvd.reducescatter(x, op=hvd.Sum, name="rsg")
        (y.sum() * (rank + 1)).backward()
        # row block b of x contributes to rank b's shard; weight (b+1)
        exp = torch.tensor([[1.0] * 3] * 2 + [[2.0] * 3] * 2)
        assert torch.allclose(x.grad, exp), x.grad
    """, timeout=240)


def test_expert_parallel_routing_np4():
    """Dogfood the EP/Ulysses primitive combo (process sets +
    alltoall are the building blocks): tokens routed to per-rank experts by
    alltoall-v, processed, and returned by the reverse alltoall — gradients
    flow end to end."""
    run_workers(4, """
        torch.manual_seed(23)
        n = size
        expert = torch.nn.Linear(4, 4)  # same init everywhere (seed)
        hvd.broadcast_parameters(expert.state_dict(), root_rank=0)
        # 8 tokens per rank, each assigned a destination expert
        g = torch.Generator().manual_seed(100 + rank)
        tokens = torch.randn(8, 4, generator=g).requires_grad_(True)
        dest = torch.randint(0, n, (8,),
                             generator=torch.Generator().manual_seed(rank))
        order = torch.argsort(dest)
        routed = tokens[order]
        splits = torch.bincount(dest, minlength=n)
        inbox, recv_splits = hvd.alltoall(routed, splits=splits,
                                          name="ep.fwd")
        hidden = torch.nn.functional.gelu(expert(inbox))
        outbox, _ = hvd.alltoall(hidden, splits=recv_splits, name="ep.bwd")
        assert outbox.shape == routed.shape
        loss = outbox.pow(2).sum()
        loss.backward()
        assert tokens.grad is not None
        assert tokens.grad.abs().sum() > 0
        # expert grads exist and are LOCAL (per-rank expert => not averaged)
        assert expert.weight.grad is not None
    """, timeout=300)


def test_timeline_stop_is_synchronous_np2():
    """Round-2 regression: stop_timeline() must return with the file
    already complete, valid JSON (previously the writer finalized
    asynchronously, so an immediate read raced the flush and one rank's
    crash cascaded into peer connection resets)."""
    import json as _json
    run_workers(2, """
        import json
        for cycle in range(8):
            base = f"/tmp/hvd_tl_sync_{rank}_{cycle}.json"
            hvd.start_timeline(base if rank == 0 else base[:-5])
            for i in range(8):
                hvd.allreduce(torch.ones(128), average=False, name=f"t{i}")
            hvd.stop_timeline()
            mine = base if rank == 0 else base[:-5] + ".1"
            data = json.load(open(mine))
            assert isinstance(data, list) and data, (cycle, mine)
            import os
            os.unlink(mine)
    """, timeout=240)
