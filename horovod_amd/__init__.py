"""horovod_amd: an MI355X-native distributed deep-learning framework with
Horovod's capabilities (ring-allreduce data parallelism, elastic training,
a horovodrun-compatible launcher) built on PyTorch-ROCm, hand-written
CDNA4/HIP kernels and RCCL over xGMI.

Structural reference: horovod/horovod v0.28.1.  This is a
from-scratch design, not a port: single GPU data plane (RCCL), TCP star
control plane (no MPI/Gloo), torch-only front-end.
"""
# torch must load before the native extension: _core.so links libtorch, and
# letting the dynamic loader pull libtorch in WITHOUT the python module's
# own initialization corrupts interpreter teardown (glibc fastbin abort at
# exit, observed under pytest).  Importing torch here guarantees ordering
# for any `import horovod_amd._core`.
import torch  # noqa: F401  (ordering dependency, see above)

__version__ = "0.1.0"


def run(*args, **kwargs):
    """In-process launcher API (reference: horovod.run)."""
    from horovod_amd.runner.interactive import run as _run
    return _run(*args, **kwargs)
