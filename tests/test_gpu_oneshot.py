"""One-shot xGMI allreduce tests that run on a SINGLE GPU.

RCCL refuses two ranks on one device, but the one-shot path (csrc/
oneshot.hip) doesn't use RCCL at all — hipIpc windows + system-scope flag
protocol work between two processes sharing device 0 (validated by
examples/ipc_probe.hip).  So the full protocol — handle exchange over the
star, pack-to-staging, ready/consumed sequencing, slot reuse, the reduce
kernel — is testable on the 1-GPU CI box.  Every op here must stay
one-shot-ELIGIBLE (float dtype, under HOROVOD_ONESHOT_THRESHOLD): an
ineligible op would fall back to RCCL and abort on the shared device.

Real multi-GPU xGMI coverage: test_gpu_multi.py::test_multi_oneshot.
"""
import pytest
import torch

from tests.parallel_util import run_workers

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")

ENV = {"HOROVOD_ONESHOT_ALLREDUCE": "1"}

PRELUDE = """
        torch.cuda.set_device(hvd.local_rank() % torch.cuda.device_count())
        dev = torch.device("cuda",
                           hvd.local_rank() % torch.cuda.device_count())
"""


@requires_gpu
def test_oneshot_allreduce_repeat():
    """Repeated ops cycle through staging slots (seq > nslots exercises the
    consumed-wait) and stay exact."""
    run_workers(2, PRELUDE + """
        for i in range(12):
            t = torch.full((1024,), float(rank + 1 + i), device=dev)
            out = hvd.allreduce(t, average=False, name=f"os.{i}")
            exp = torch.full_like(t, float(3 + 2 * i))
            assert torch.equal(out, exp), (i, out[0].item())
        from horovod_amd import _core
        assert not _core.rccl_used(), "one-shot ops must not create RCCL comms"
    """, extra_env=ENV, timeout=300)


@requires_gpu
def test_oneshot_grouped_and_ops():
    """Fused (grouped) buckets, min/max/product, odd sizes."""
    run_workers(2, PRELUDE + """
        ts = [torch.arange(k, dtype=torch.float32, device=dev) * (rank + 1)
              for k in (1, 17, 1000, 4097)]
        outs = hvd.grouped_allreduce(ts, average=False, name="osg")
        for k, o in zip((1, 17, 1000, 4097), outs):
            exp = torch.arange(k, dtype=torch.float32, device=dev) * 3
            assert torch.equal(o, exp), k
        a = torch.tensor([1.0, 5.0], device=dev) if rank == 0 else \\
            torch.tensor([3.0, 2.0], device=dev)
        mn = hvd.allreduce(a, op=hvd.Min, name="osmn")
        mx = hvd.allreduce(a, op=hvd.Max, name="osmx")
        pr = hvd.allreduce(a, op=hvd.Product, name="ospr")
        assert torch.equal(mn.cpu(), torch.tensor([1.0, 2.0])), mn
        assert torch.equal(mx.cpu(), torch.tensor([3.0, 5.0])), mx
        assert torch.equal(pr.cpu(), torch.tensor([3.0, 10.0])), pr
    """, extra_env=ENV, timeout=300)


@requires_gpu
def test_oneshot_average_scale_bf16():
    """Average (postscale in unpack), pre/postscale, bf16 wire."""
    run_workers(2, PRELUDE + """
        t = torch.full((640,), float(rank + 1), device=dev)
        avg = hvd.allreduce(t, name="osavg")
        assert torch.allclose(avg, torch.full_like(t, 1.5)), avg[0]
        sc = hvd.allreduce(t, average=False, prescale_factor=2.0,
                           postscale_factor=0.25, name="ossc")
        assert torch.allclose(sc, torch.full_like(t, 1.5)), sc[0]
        o = hvd.allreduce(t, average=False, name="osbf",
                          compression=hvd.Compression.bf16)
        assert torch.allclose(o, torch.full_like(t, 3.0), rtol=1e-2), o[0]
        tb = torch.full((320,), float(rank + 1), device=dev,
                        dtype=torch.bfloat16)
        ob = hvd.allreduce(tb, average=False, name="osbd")
        assert torch.allclose(ob.float(), torch.full((320,), 3.0,
                                                     device=dev)), ob[0]
    """, extra_env=ENV, timeout=300)


@requires_gpu
def test_oneshot_numerics_random():
    """Random payloads: bit-identical results on both ranks vs a local
    fp32 oracle (closed-form seeds)."""
    run_workers(2, PRELUDE + """
        for trial in range(4):
            gs = [torch.Generator().manual_seed(7000 + trial * 10 + r)
                  for r in range(size)]
            vals = [torch.randn(3333, generator=g) for g in gs]
            t = vals[rank].to(dev)
            out = hvd.allreduce(t, average=False, name=f"osr.{trial}")
            exp = (vals[0] + vals[1]).to(dev)
            assert torch.allclose(out, exp, atol=1e-6), \\
                (out - exp).abs().max().item()
    """, extra_env=ENV, timeout=300)


@requires_gpu
def test_oneshot_peer_death_unwedges_spin():
    """Kill one rank between ops: the survivor's spin-wait kernel must exit
    via the host-raised abort word (no wedged GPU), and its pending op must
    fail with HorovodInternalError."""
    import os
    import subprocess
    import sys
    import time
    from horovod_amd.runner.launch import find_free_port, slot_env
    from tests.parallel_util import REPO

    # The victim negotiates a burst of async ops (so the survivor's pack/
    # reduce kernels are enqueued and some spin on the victim's never-set
    # ready flags), then dies without running/synchronizing them.
    victim = (
        "import torch, horovod_amd.torch as hvd, os, time\n"
        "hvd.init(); torch.cuda.set_device(0)\n"
        "hvd.allreduce(torch.ones(256, device='cuda'), name='osf0')\n"
        "hs = [hvd.allreduce_async(torch.ones(256, device='cuda'),"
        " name=f'osf{i}') for i in range(1, 16)]\n"
        "time.sleep(0.3)  # let negotiation finish; do NOT synchronize\n"
        "os._exit(9)\n"
    )
    survivor = (
        "import torch, horovod_amd.torch as hvd\n"
        "from horovod_amd.common.exceptions import HorovodInternalError\n"
        "hvd.init(); torch.cuda.set_device(0)\n"
        "hvd.allreduce(torch.ones(256, device='cuda'), name='osf0')\n"
        "hs = [hvd.allreduce_async(torch.ones(256, device='cuda'),"
        " name=f'osf{i}') for i in range(1, 16)]\n"
        "try:\n"
        "    for h in hs:\n"
        "        hvd.synchronize(h)\n"
        "    for i in range(16, 40):\n"
        "        hvd.allreduce(torch.ones(256, device='cuda'),"
        " name=f'osf{i}')\n"
        "    print('UNEXPECTED_SUCCESS')\n"
        "except (HorovodInternalError, RuntimeError):\n"
        "    print('GOT_ERROR')\n"
        "import torch as t\n"
        "t.cuda.synchronize()\n"  # GPU must not be wedged
        "print('GPU_RESPONSIVE')\n"
    )
    port = find_free_port()
    procs = []
    for r, body in enumerate([survivor, victim]):
        env = slot_env(r, 2, r, 2, 0, 1, "127.0.0.1", port)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        env["HOROVOD_ONESHOT_ALLREDUCE"] = "1"
        env["HOROVOD_SHUTDOWN_GRACE_SECONDS"] = "2"
        procs.append(subprocess.Popen([sys.executable, "-c", body], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    try:
        out0, _ = procs[0].communicate(timeout=150)
    except subprocess.TimeoutExpired:
        for q in procs:
            q.kill()
        raise AssertionError("survivor hung: oneshot spin not aborted")
    procs[1].wait(timeout=30)
    assert "GOT_ERROR" in out0, out0[-600:]
    assert "GPU_RESPONSIVE" in out0, out0[-600:]
