"""Process bootstrap: env-derived topology + native core init.

Reference: horovod/common/basics.py (HorovodBasics).  Differences by design:
no MPI — rank/size come from the launcher environment (our `horovodrun`
equivalent, or torchrun's RANK/WORLD_SIZE), and the control plane is the
in-core TCP star whose rendezvous address is derived from the same env.
"""
import atexit
import os

from horovod_amd import _core


def _env_int(names, default):
    for n in names:
        v = os.environ.get(n)
        if v is not None:
            try:
                return int(v)
            except ValueError:
                pass
    return default


def _env_str(names, default):
    for n in names:
        v = os.environ.get(n)
        if v:
            return v
    return default


class HorovodBasics:
    def __init__(self):
        self._initialized_here = False

    def init(self, comm=None, process_sets=None):
        """Initialize horovod_amd.

        `comm` may be a list of ranks (subset init is not supported — pass
        None) and is accepted for API compatibility.  `process_sets` is a list
        of ProcessSet objects (or the string "dynamic") registered at startup.
        """
        if _core.is_initialized():
            return
        rank = _env_int(["HOROVOD_RANK", "RANK"], 0)
        size = _env_int(["HOROVOD_SIZE", "WORLD_SIZE"], 1)
        local_rank = _env_int(["HOROVOD_LOCAL_RANK", "LOCAL_RANK"], rank)
        local_size = _env_int(["HOROVOD_LOCAL_SIZE", "LOCAL_WORLD_SIZE"], size)
        cross_rank = _env_int(["HOROVOD_CROSS_RANK"], rank // max(local_size, 1))
        cross_size = _env_int(["HOROVOD_CROSS_SIZE"],
                              max(1, size // max(local_size, 1)))
        addr = _env_str(["HOROVOD_CONTROLLER_ADDR", "MASTER_ADDR"], "127.0.0.1")
        # offset from MASTER_PORT so we never collide with torchrun's TCPStore
        base_port = _env_int(["MASTER_PORT"], 29500)
        port = _env_int(["HOROVOD_CONTROLLER_PORT"], base_port + 73)

        fusion_mb = float(os.environ.get("HOROVOD_FUSION_THRESHOLD",
                                         64 * 1024 * 1024))
        cycle_ms = float(os.environ.get("HOROVOD_CYCLE_TIME", 1.0))
        cache_cap = _env_int(["HOROVOD_CACHE_CAPACITY"], 1024)
        stall_sec = float(os.environ.get("HOROVOD_STALL_CHECK_TIME_SECONDS", 60.0))
        stall_shutdown = float(os.environ.get(
            "HOROVOD_STALL_SHUTDOWN_TIME_SECONDS", 0.0))
        timeline = bool(os.environ.get("HOROVOD_TIMELINE"))

        _core.init(rank=rank, size=size, local_rank=local_rank,
                   local_size=local_size, cross_rank=cross_rank,
                   cross_size=cross_size, addr=addr, port=port,
                   fusion_threshold=int(fusion_mb), cycle_time_ms=cycle_ms,
                   cache_capacity=cache_cap, stall_warning_sec=stall_sec,
                   stall_shutdown_sec=stall_shutdown, timeline=timeline)
        self._initialized_here = True
        atexit.register(self.shutdown)

        if process_sets:
            from horovod_amd.common import process_sets as ps_mod
            if not isinstance(process_sets, str):
                for ps in process_sets:
                    ps_mod.add_process_set(ps)

    def shutdown(self):
        if _core.is_initialized():
            _core.shutdown()

    def is_initialized(self):
        return _core.is_initialized()

    # topology queries -------------------------------------------------------
    def rank(self):
        self._check()
        return _core.rank()

    def size(self):
        self._check()
        return _core.size()

    def local_rank(self):
        self._check()
        return _core.local_rank()

    def local_size(self):
        self._check()
        return _core.local_size()

    def cross_rank(self):
        self._check()
        return _core.cross_rank()

    def cross_size(self):
        self._check()
        return _core.cross_size()

    def is_homogeneous(self):
        return True

    # capability probes (reference: basics.py mpi_built etc.) ----------------
    def mpi_threads_supported(self):
        return False

    def mpi_enabled(self):
        return False

    def mpi_built(self):
        return False

    def gloo_enabled(self):
        # the TCP controller fills gloo's role; scripts probing for a
        # non-MPI controller should take the gloo path
        return True

    def gloo_built(self):
        return True

    def nccl_built(self):
        return True  # RCCL

    def ddl_built(self):
        return False

    def ccl_built(self):
        return False

    def cuda_built(self):
        return False

    def rocm_built(self):
        return True

    def _check(self):
        if not _core.is_initialized():
            raise ValueError(
                "Horovod has not been initialized; use hvd.init().")
