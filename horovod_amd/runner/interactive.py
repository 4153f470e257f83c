"""In-process launcher: horovod_amd.run(fn, args=(), np=N, ...) executes
`fn` on np local processes and returns the per-rank results.

Reference: horovod.run / runner/__init__.py (interactive API used by
test_interactiverun.py and the Spark layer).
"""
import os
import subprocess
import sys
import tempfile

import cloudpickle

from horovod_amd.runner.launch import find_free_port, slot_env

_WORKER_SRC = r"""
import base64, pickle, sys
import cloudpickle

payload_file, out_file = sys.argv[1], sys.argv[2]
with open(payload_file, 'rb') as f:
    fn, args, kwargs = cloudpickle.load(f)
result = fn(*args, **kwargs)
with open(out_file, 'wb') as f:
    cloudpickle.dump(result, f)
"""


def run(fn, args=(), kwargs=None, np=1, use_gloo=None, use_mpi=None,
        hosts=None, env=None, verbose=False, **_compat):
    """Run `fn(*args, **kwargs)` on np processes; returns [result_rank0, ...].

    use_gloo/use_mpi accepted for API compatibility (always the TCP/gloo-style
    controller here)."""
    kwargs = kwargs or {}
    with tempfile.TemporaryDirectory() as tmp:
        payload = os.path.join(tmp, "fn.pkl")
        with open(payload, "wb") as f:
            cloudpickle.dump((fn, args, kwargs), f)
        worker = os.path.join(tmp, "worker.py")
        with open(worker, "w") as f:
            f.write(_WORKER_SRC)
        outs = [os.path.join(tmp, f"out.{r}.pkl") for r in range(np)]
        port = find_free_port()
        procs = []
        for r in range(np):
            e = slot_env(r, np, r, np, 0, 1, "127.0.0.1", port,
                         dict(env or os.environ))
            repo_root = os.path.dirname(os.path.dirname(os.path.dirname(
                os.path.abspath(__file__))))
            e["PYTHONPATH"] = repo_root + os.pathsep + e.get("PYTHONPATH", "")
            procs.append(subprocess.Popen(
                [sys.executable, worker, payload, outs[r]], env=e))
        codes = [p.wait() for p in procs]
        if any(c != 0 for c in codes):
            raise RuntimeError(f"horovod_amd.run: worker exit codes {codes}")
        results = []
        for r in range(np):
            with open(outs[r], "rb") as f:
                results.append(cloudpickle.load(f))
        return results
