"""Exceptions (reference: horovod/common/exceptions.py:18-32)."""


class HorovodInternalError(RuntimeError):
    """Internal error raised when a collective routine fails.

    Elastic mode catches this, restores state from the last commit and
    re-initializes from a fresh rendezvous.
    """


class HostsUpdatedInterrupt(RuntimeError):
    """Raised when the host set changed (elastic); current state is kept."""

    def __init__(self, skip_sync=False):
        super().__init__()
        self.skip_sync = skip_sync
