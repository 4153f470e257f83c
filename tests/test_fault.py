"""Failure-detection tests: a dead peer must surface as HorovodInternalError
on the survivors within a bounded time — the signal elastic mode recovers
from (reference: SURVEY.md §5 failure detection)."""
import os
import subprocess
import sys
import time

from tests.parallel_util import REPO


def test_peer_death_raises(tmp_path):
    from horovod_amd.runner.launch import find_free_port, slot_env

    victim = (
        "import torch, horovod_amd.torch as hvd, os, sys\n"
        "hvd.init()\n"
        "sys.exit(3)  # die before any collective\n"
    )
    survivor = (
        "import torch, horovod_amd.torch as hvd, sys\n"
        "hvd.init()\n"
        "try:\n"
        "    out = hvd.allreduce(torch.ones(4), average=False, name='x')\n"
        "    print('UNEXPECTED_SUCCESS')\n"
        "except RuntimeError as e:\n"
        "    print('GOT_ERROR', type(e).__name__, str(e)[:80])\n"
    )
    port = find_free_port()
    env0 = slot_env(0, 2, 0, 2, 0, 1, "127.0.0.1", port)
    env1 = slot_env(1, 2, 1, 2, 0, 1, "127.0.0.1", port)
    for e in (env0, env1):
        e["PYTHONPATH"] = REPO + os.pathsep + e.get("PYTHONPATH", "")
        e["HOROVOD_SHUTDOWN_GRACE_SECONDS"] = "2"
    p0 = subprocess.Popen([sys.executable, "-c", survivor], env=env0,
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                          text=True)
    p1 = subprocess.Popen([sys.executable, "-c", victim], env=env1)
    t0 = time.time()
    out0, _ = p0.communicate(timeout=120)
    p1.wait(timeout=30)
    elapsed = time.time() - t0
    assert "GOT_ERROR" in out0, out0
    assert elapsed < 60, f"survivor took {elapsed}s to notice the dead peer"


def test_stall_inspector_warns(tmp_path):
    """One rank never submits; the coordinator must print a stall warning
    naming the missing rank (reference: test_stall.py)."""
    import subprocess
    import sys
    from horovod_amd.runner.launch import find_free_port, slot_env

    submitter = (
        "import torch, horovod_amd.torch as hvd, time\n"
        "hvd.init()\n"
        "h = hvd.allreduce_async(torch.ones(4), average=False, name='s')\n"
        "time.sleep(6)\n"
    )
    laggard = (
        "import torch, horovod_amd.torch as hvd, time\n"
        "hvd.init()\n"
        "time.sleep(6)\n"
    )
    port = find_free_port()
    env = {}
    procs = []
    outs = []
    for r, src in enumerate([submitter, laggard]):
        e = slot_env(r, 2, r, 2, 0, 1, "127.0.0.1", port)
        e["PYTHONPATH"] = REPO + os.pathsep + e.get("PYTHONPATH", "")
        e["HOROVOD_STALL_CHECK_TIME_SECONDS"] = "2"
        procs.append(subprocess.Popen([sys.executable, "-c", src], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    for p in procs:
        out, _ = p.communicate(timeout=60)
        outs.append(out)
    assert "stalled" in outs[0], outs
    assert "waiting on ranks: 1" in outs[0], outs[0]
