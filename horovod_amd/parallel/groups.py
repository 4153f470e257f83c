"""Parallelism-group helpers built on process sets.

The reference exposes process sets as the building block users combine with
alltoall/reducescatter for TP/SP-style schemes; these
helpers construct the standard 2-D (dp x tp) grid of process sets so model
code can pass `process_set=` per collective.

On one 8-GPU MI355X node the xGMI mesh is fully connected (7 links/GPU), so
any grid slicing has full-bandwidth point-to-point paths — pick tp degree by
model shape, not topology.
"""
from horovod_amd.common.process_sets import ProcessSet, add_process_set
from horovod_amd.torch.mpi_ops import rank, size


def grid_process_sets(tp_degree):
    """Build tensor-parallel and data-parallel process sets for a
    (size/tp) x tp grid.  Must be called identically on every rank.

    Returns (my_tp_set, my_dp_set, all_tp_sets, all_dp_sets)."""
    world = size()
    if world % tp_degree != 0:
        raise ValueError(f"world size {world} not divisible by tp={tp_degree}")
    dp_degree = world // tp_degree
    tp_sets = []
    for d in range(dp_degree):
        tp_sets.append(add_process_set(
            ProcessSet(range(d * tp_degree, (d + 1) * tp_degree))))
    dp_sets = []
    for t in range(tp_degree):
        dp_sets.append(add_process_set(
            ProcessSet(range(t, world, tp_degree))))
    me = rank()
    my_tp = tp_sets[me // tp_degree]
    my_dp = dp_sets[me % tp_degree]
    return my_tp, my_dp, tp_sets, dp_sets
