"""Build/system invariants: the in-tree native extension is what loads and
exposes the full binding surface."""
import os

import horovod_amd._core as core
from tests.parallel_util import REPO


def test_in_tree_extension():
    assert core.__file__ == os.path.join(REPO, "horovod_amd", "_core.so"), \
        core.__file__


def test_binding_surface():
    expected = [
        "init", "shutdown", "is_initialized", "rank", "size", "local_rank",
        "local_size", "cross_rank", "cross_size", "rccl_used",
        "allreduce_async", "allgather_async", "broadcast_async",
        "alltoall_async", "reducescatter_async", "join_async",
        "barrier_async", "poll", "wait", "flush", "add_process_set",
        "remove_process_set", "process_set_ranks", "set_fusion_threshold",
        "get_fusion_threshold", "set_cycle_time_ms", "get_cycle_time_ms",
        "start_timeline", "stop_timeline", "fused_sgd_step",
        "adasum_combine_", "fused_bn_relu_forward", "fused_bn_relu_backward",
    ]
    missing = [n for n in expected if not hasattr(core, n)]
    assert not missing, missing


def test_gfx950_code_object():
    """The shipped .so must carry a gfx950 device image (not a foreign
    arch)."""
    with open(core.__file__, "rb") as f:
        blob = f.read()
    assert b"gfx950" in blob, "no gfx950 code object in _core.so"
