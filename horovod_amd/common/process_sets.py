"""Process sets: concurrent subgroup communicators.

Reference: horovod/common/process_sets.py:18-163.  The native side keeps a
single global controller that negotiates every set (controller.h), so
registration is a local call that must be made identically on ALL ranks (the
same requirement the reference documents for dynamic sets).
"""
from horovod_amd import _core


class ProcessSet:
    """An object capturing a subset of ranks usable in any hvd.* collective."""

    process_set_id = None

    def __init__(self, ranks_or_comm):
        self.ranks = sorted(int(r) for r in ranks_or_comm)

    def size(self):
        if self.process_set_id is None:
            return None
        return len(self.ranks)

    def rank(self):
        """Set-local rank of this process, or None if not a member."""
        if self.process_set_id is None:
            return None
        me = _core.rank()
        return self.ranks.index(me) if me in self.ranks else None

    def included(self):
        return _core.rank() in self.ranks

    def __str__(self):
        return f"ProcessSet(process_set_id={self.process_set_id}, ranks={self.ranks})"


class _GlobalProcessSet(ProcessSet):
    process_set_id = 0

    def __init__(self):
        self.ranks = []

    def size(self):
        return _core.size()

    def rank(self):
        return _core.rank()

    def included(self):
        return True


global_process_set = _GlobalProcessSet()


_known_sets = []


def add_process_set(process_set):
    """Register a new process set.  Must be called on every rank with the
    same arguments, in the same order (collective registration)."""
    if isinstance(process_set, (list, tuple)):
        process_set = ProcessSet(process_set)
    if process_set.process_set_id is not None:
        raise ValueError("process set already registered")
    process_set.process_set_id = _core.add_process_set(process_set.ranks)
    _known_sets.append(process_set)
    # registration is collective (must be called identically on every rank,
    # reference: horovod_add_process_set operations.cc:1262-1328); barrier so
    # no rank races ahead and uses the set before peers registered it
    h = _core.barrier_async(0)
    _core.wait(h)
    return process_set


def remove_process_set(process_set):
    if process_set.process_set_id in (None, 0):
        return False
    _core.remove_process_set(process_set.process_set_id)
    process_set.process_set_id = None
    if process_set in _known_sets:
        _known_sets.remove(process_set)
    return True


def _registered_sets():
    """All ProcessSet objects registered in this process (global set
    included) — used by process_set_included and tooling."""
    return list(_known_sets)
