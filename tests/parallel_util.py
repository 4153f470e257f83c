"""Multi-process test harness: run a pytest-style worker function under the
real launcher's slot-env protocol on localhost (reference test strategy:
SURVEY.md §4 tier 2 — every parallel test file executes under a real
N-process launch)."""
import os
import subprocess
import sys
import textwrap

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_workers(np, body, timeout=300, extra_env=None):
    """Run `body` (python source; has `hvd`, `torch`, `rank`, `size` in
    scope) in np processes.  Raises on nonzero exits; returns list of stdout
    strings by rank."""
    from horovod_amd.runner.launch import find_free_port, slot_env

    prelude = ("import sys\n"
               "import torch\n"
               "import horovod_amd.torch as hvd\n"
               "hvd.init()\n"
               "rank, size = hvd.rank(), hvd.size()\n")
    epilogue = ("\nhvd.shutdown()\n"
                "print('WORKER_OK', rank)\n")
    script = prelude + textwrap.dedent(body) + epilogue
    port = find_free_port()
    procs = []
    for r in range(np):
        env = slot_env(r, np, r, np, 0, 1, "127.0.0.1", port)
        env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
        if extra_env:
            env.update({k: str(v) for k, v in extra_env.items()})
        procs.append(subprocess.Popen([sys.executable, "-c", script], env=env,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT, text=True))
    outs = []
    fail = None
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise AssertionError(f"rank {r} timed out")
        outs.append(out)
        if p.returncode != 0 or "WORKER_OK" not in out:
            fail = (r, p.returncode, out)
    if fail:
        raise AssertionError(
            f"rank {fail[0]} failed (exit {fail[1]}):\n{fail[2]}")
    return outs
