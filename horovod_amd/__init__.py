"""horovod_amd: an MI355X-native distributed deep-learning framework with
Horovod's capabilities (ring-allreduce data parallelism, elastic training,
a horovodrun-compatible launcher) built on PyTorch-ROCm, hand-written
CDNA4/HIP kernels and RCCL over xGMI.

Structural reference: horovod/horovod v0.28.1.  This is a
from-scratch design, not a port: single GPU data plane (RCCL), TCP star
control plane (no MPI/Gloo), torch-only front-end.
"""
__version__ = "0.1.0"


def run(*args, **kwargs):
    """In-process launcher API (reference: horovod.run)."""
    from horovod_amd.runner.interactive import run as _run
    return _run(*args, **kwargs)
