"""Elastic fault-tolerance core: State objects + the retry loop.

Reference: horovod/common/elastic.py:26-175 (State/ObjectState with
save/restore/sync and commit-with-host-update-check; run_fn catching
HorovodInternalError -> restore+reinit and HostsUpdatedInterrupt -> reinit).

Worker-side mechanics on this stack: the elastic driver
(horovod_amd/runner/elastic_driver.py) publishes versioned rank assignments
over an HTTP rendezvous; a worker reset shuts the native core down, polls the
rendezvous for its next assignment, rewrites the HOROVOD_* env and
re-initializes the TCP controller + RCCL comms.
"""
import functools
import json
import os
import threading
import time
import urllib.request

from horovod_amd.common.exceptions import (HorovodInternalError,
                                           HostsUpdatedInterrupt)


class _WorkerNotificationManager:
    """Polls the rendezvous /version endpoint; flags host updates so
    State.commit() can raise HostsUpdatedInterrupt (reference:
    WorkerNotificationService push, runner/elastic/worker.py:46-119 — we poll
    instead of push: simpler and NAT-proof)."""

    def __init__(self):
        self._thread = None
        self._stop = threading.Event()
        self._updated = threading.Event()
        self._known_version = int(os.environ.get("HOROVOD_ELASTIC_VERSION", 0))

    def init(self):
        if self._thread is not None or not rendezvous_addr():
            return
        self._thread = threading.Thread(target=self._poll, daemon=True)
        self._thread.start()

    def _poll(self):
        while not self._stop.is_set():
            try:
                v = rendezvous_get("version").get("version", self._known_version)
                if v > self._known_version:
                    self._updated.set()
                    self._interrupt_if_displaced()
            except Exception:
                pass
            time.sleep(0.5)

    def _interrupt_if_displaced(self):
        """Prompt scale-down (round-2): a worker dropped from the new
        assignment must stop within ~1 s regardless of commit cadence
        (reference pushes notifications, worker.py:46-119; round-1 only
        noticed at the next commit()).  Interrupting the native core fails
        its next collective with HorovodInternalError, so the retry loop
        resets immediately and the rendezvous answers `terminate`."""
        try:
            wid = os.environ.get("HOROVOD_WORKER_ID")
            if not wid:
                return
            a = rendezvous_get("assignment", worker_id=wid,
                               after=self._known_version)
            if a.get("terminate"):
                from horovod_amd import _core
                _core.interrupt("displaced by elastic host update")
        except Exception:
            pass

    def host_updates_available(self):
        return self._updated.is_set()

    def acknowledge(self, version):
        self._known_version = version
        self._updated.clear()


notification_manager = _WorkerNotificationManager()


def rendezvous_addr():
    a = os.environ.get("HOROVOD_RENDEZVOUS_ADDR")
    p = os.environ.get("HOROVOD_RENDEZVOUS_PORT")
    return (a, int(p)) if a and p else None


def rendezvous_get(path, **params):
    addr = rendezvous_addr()
    if not addr:
        raise RuntimeError("no rendezvous configured (HOROVOD_RENDEZVOUS_*)")
    qs = "&".join(f"{k}={v}" for k, v in params.items())
    url = f"http://{addr[0]}:{addr[1]}/{path}" + (f"?{qs}" if qs else "")
    with urllib.request.urlopen(url, timeout=10) as r:
        return json.loads(r.read().decode())


def _apply_assignment(a):
    os.environ.update({
        "HOROVOD_RANK": str(a["rank"]),
        "HOROVOD_SIZE": str(a["size"]),
        "HOROVOD_LOCAL_RANK": str(a["local_rank"]),
        "HOROVOD_LOCAL_SIZE": str(a["local_size"]),
        "HOROVOD_CROSS_RANK": str(a["cross_rank"]),
        "HOROVOD_CROSS_SIZE": str(a["cross_size"]),
        "HOROVOD_CONTROLLER_ADDR": a["controller_addr"],
        "HOROVOD_CONTROLLER_PORT": str(a["controller_port"]),
        "HOROVOD_ELASTIC_VERSION": str(a["version"]),
    })


def _rendezvous_reset(timeout=600):
    """Shut down the core and re-init from the next rendezvous assignment."""
    from horovod_amd import _core
    if _core.is_initialized():
        _core.shutdown()
    wid = os.environ["HOROVOD_WORKER_ID"]
    after = int(os.environ.get("HOROVOD_ELASTIC_VERSION", 0))
    deadline = time.time() + timeout
    while True:
        try:
            a = rendezvous_get("assignment", worker_id=wid, after=after)
            if not a.get("pending"):
                break
        except Exception:
            pass
        if time.time() > deadline:
            raise RuntimeError("elastic rendezvous timed out")
        time.sleep(0.5)
    if a.get("terminate"):
        raise SystemExit(0)
    _apply_assignment(a)
    notification_manager.acknowledge(a["version"])
    from horovod_amd.common.basics import HorovodBasics
    HorovodBasics().init()
    return a


class State:
    """Mutable training state with commit/restore/sync semantics
    (reference: common/elastic.py:26-148)."""

    def __init__(self, bcast_object, get_rank):
        self._bcast_object = bcast_object
        self._rank = get_rank
        self._reset_callbacks = []

    def register_reset_callbacks(self, callbacks):
        self._reset_callbacks.extend(callbacks)

    def on_reset(self):
        self.reset()
        for cb in self._reset_callbacks:
            cb()

    def commit(self):
        self.save()
        self.check_host_updates()

    def check_host_updates(self):
        if notification_manager.host_updates_available():
            raise HostsUpdatedInterrupt(skip_sync=False)

    def save(self):
        raise NotImplementedError()

    def restore(self):
        raise NotImplementedError()

    def sync(self):
        raise NotImplementedError()

    def reset(self):
        pass


class ObjectState(State):
    """State backed by an in-memory dict of attributes, synced via
    broadcast_object (reference: common/elastic.py ObjectState)."""

    def __init__(self, bcast_object, get_rank, **kwargs):
        self._saved_state = kwargs
        for k, v in kwargs.items():
            setattr(self, k, v)
        super().__init__(bcast_object, get_rank)

    def save(self):
        new_state = {}
        for k in self._saved_state.keys():
            new_state[k] = getattr(self, k)
        self._saved_state = new_state

    def restore(self):
        for k, v in self._saved_state.items():
            setattr(self, k, v)

    def sync(self):
        if self._saved_state:
            synced = self._bcast_object(self._saved_state)
            if self._rank() != 0:
                self._saved_state = synced
                self.restore()


def run_fn(func, reset):
    """The elastic retry loop (reference: common/elastic.py:151-175)."""
    @functools.wraps(func)
    def wrapper(state, *args, **kwargs):
        notification_manager.init()
        try:
            while True:
                state.sync()
                try:
                    return func(state, *args, **kwargs)
                except HorovodInternalError:
                    state.restore()
                    reset()
                    state.on_reset()
                except HostsUpdatedInterrupt:
                    reset()
                    state.on_reset()
        finally:
            pass
    return wrapper
