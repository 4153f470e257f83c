"""Elastic driver: discovery, worker lifecycle, rank assignment, rendezvous.

Reference: horovod/runner/elastic/{driver.py,discovery.py,registration.py,
rendezvous.py} — discovery thread polling a host script, blacklist with
cooldown, rank assignment preserving host order, worker exit handling, HTTP
rendezvous feeding new ranks.

This driver publishes versioned assignments over a small HTTP server
(ThreadingHTTPServer); workers poll it on reset (see common/elastic.py).
"""
import json
import os
import subprocess
import sys
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, urlparse

from horovod_amd.runner.launch import find_free_port


class HostDiscovery:
    def find_available_hosts_and_slots(self):
        raise NotImplementedError()


class HostDiscoveryScript(HostDiscovery):
    """Runs an executable that prints `host:slots` (or `host slots=N`) lines
    (reference: runner/elastic/discovery.py:232-263)."""

    def __init__(self, script, default_slots=1):
        self.script = script
        self.default_slots = default_slots

    def find_available_hosts_and_slots(self):
        out = subprocess.run([self.script], capture_output=True, text=True,
                             timeout=30)
        hosts = {}
        if out.returncode != 0:
            return hosts
        for line in out.stdout.splitlines():
            line = line.strip()
            if not line:
                continue
            if ":" in line:
                h, s = line.rsplit(":", 1)
                hosts[h] = int(s)
            else:
                parts = line.split()
                slots = self.default_slots
                for p in parts[1:]:
                    if p.startswith("slots="):
                        slots = int(p.split("=")[1])
                hosts[parts[0]] = slots
        return hosts


class FixedHosts(HostDiscovery):
    """In-memory discovery for tests (reference: discovery.py:265-274)."""

    def __init__(self, hosts):
        self._hosts = dict(hosts)

    def set(self, hosts):
        self._hosts = dict(hosts)

    def find_available_hosts_and_slots(self):
        return dict(self._hosts)


class _RendezvousHandler(BaseHTTPRequestHandler):
    driver = None

    def log_message(self, *args):
        pass

    def do_GET(self):
        parsed = urlparse(self.path)
        qs = {k: v[0] for k, v in parse_qs(parsed.query).items()}
        d = self.driver
        if parsed.path == "/version":
            body = {"version": d.version}
        elif parsed.path == "/assignment":
            body = d.get_assignment(qs.get("worker_id"),
                                    int(qs.get("after", -1)))
        else:
            self.send_response(404)
            self.end_headers()
            return
        data = json.dumps(body).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)


class ElasticDriver:
    def __init__(self, discovery, command, env=None, min_np=1, max_np=None,
                 reset_limit=None, cooldown=30.0, discovery_interval=1.0):
        self.discovery = discovery
        self.command = command
        self.env = dict(env or os.environ)
        self.min_np = min_np
        self.max_np = max_np
        self.reset_limit = reset_limit
        self.cooldown = cooldown
        self.discovery_interval = discovery_interval

        self.version = 0
        self.reset_count = 0
        self._lock = threading.RLock()
        self._assignments = {}        # worker_id -> assignment dict
        self._workers = {}            # worker_id -> Popen (local only)
        self._active_hosts = {}       # host -> slots
        self._blacklist = {}          # host -> blacklist timestamp
        self._shutdown = threading.Event()
        self._result = None
        self._result_event = threading.Event()

        self._server = ThreadingHTTPServer(("0.0.0.0", 0), _RendezvousHandler)
        _RendezvousHandler.driver = self
        self.rendezvous_port = self._server.server_address[1]

    # -- rendezvous --------------------------------------------------------
    def get_assignment(self, worker_id, after):
        with self._lock:
            if self.version <= after:
                return {"pending": True}
            a = self._assignments.get(worker_id)
            if a is None:
                return {"terminate": True, "version": self.version}
            return a

    # -- lifecycle ---------------------------------------------------------
    def start(self):
        threading.Thread(target=self._server.serve_forever,
                         daemon=True).start()
        hosts = self._discover()
        if not hosts:
            raise RuntimeError("elastic: no hosts discovered")
        self._update_assignments(hosts)
        threading.Thread(target=self._discovery_loop, daemon=True).start()

    def wait_for_result(self, timeout=None):
        self._result_event.wait(timeout)
        return self._result

    def stop(self):
        self._shutdown.set()
        with self._lock:
            for p in self._workers.values():
                if p.poll() is None:
                    p.terminate()
        self._server.shutdown()

    # -- internals ---------------------------------------------------------
    def _discover(self):
        hosts = self.discovery.find_available_hosts_and_slots()
        now = time.time()
        # blacklist with cooldown (reference: discovery.py:33-111)
        return {h: s for h, s in hosts.items()
                if now - self._blacklist.get(h, -1e18) > self.cooldown}

    def _discovery_loop(self):
        while not self._shutdown.is_set():
            time.sleep(self.discovery_interval)
            try:
                hosts = self._discover()
            except Exception:
                continue
            with self._lock:
                if hosts and hosts != self._active_hosts:
                    self._update_assignments(hosts)
            self._reap_workers()

    def _update_assignments(self, hosts):
        """Recompute rank assignments, preserving the order of already-active
        hosts (reference: driver.py:240-283), publish a new version and spawn
        any missing workers."""
        with self._lock:
            ordered = [h for h in self._active_hosts if h in hosts]
            ordered += [h for h in hosts if h not in ordered]
            slots = [(h, i) for h in ordered for i in range(hosts[h])]
            if self.max_np:
                slots = slots[:self.max_np]
            if len(slots) < self.min_np:
                self._finish(RuntimeError(
                    f"elastic: available slots {len(slots)} < min_np "
                    f"{self.min_np}"))
                return
            self.version += 1
            self.reset_count += 1 if self.version > 1 else 0
            if self.reset_limit is not None and \
                    self.reset_count > self.reset_limit:
                self._finish(RuntimeError("elastic: reset limit reached"))
                return
            self._active_hosts = dict(hosts)
            size = len(slots)
            controller_host = slots[0][0]
            controller_port = find_free_port()
            local_sizes = {}
            for h, _ in slots:
                local_sizes[h] = local_sizes.get(h, 0) + 1
            host_index = {h: i for i, h in
                          enumerate(dict.fromkeys(h for h, _ in slots))}
            self._assignments = {}
            for rank, (h, li) in enumerate(slots):
                wid = f"{h}:{li}"
                self._assignments[wid] = {
                    "version": self.version,
                    "rank": rank,
                    "size": size,
                    "local_rank": li,
                    "local_size": local_sizes[h],
                    "cross_rank": host_index[h],
                    "cross_size": len(host_index),
                    "controller_addr": "127.0.0.1"
                    if controller_host in ("localhost", "127.0.0.1")
                    else controller_host,
                    "controller_port": controller_port,
                }
            # spawn workers that don't exist yet
            for wid, a in self._assignments.items():
                if wid not in self._workers or \
                        self._workers[wid].poll() is not None:
                    self._spawn(wid, a)

    def _spawn(self, wid, a):
        host = wid.rsplit(":", 1)[0]
        env = dict(self.env)
        env.update({
            "HOROVOD_WORKER_ID": wid,
            "HOROVOD_ELASTIC": "1",
            "HOROVOD_RENDEZVOUS_ADDR": "127.0.0.1"
            if host in ("localhost", "127.0.0.1") else _my_addr(),
            "HOROVOD_RENDEZVOUS_PORT": str(self.rendezvous_port),
            "HOROVOD_ELASTIC_VERSION": str(a["version"]),
            "HOROVOD_RANK": str(a["rank"]),
            "HOROVOD_SIZE": str(a["size"]),
            "HOROVOD_LOCAL_RANK": str(a["local_rank"]),
            "HOROVOD_LOCAL_SIZE": str(a["local_size"]),
            "HOROVOD_CROSS_RANK": str(a["cross_rank"]),
            "HOROVOD_CROSS_SIZE": str(a["cross_size"]),
            "HOROVOD_CONTROLLER_ADDR": a["controller_addr"],
            "HOROVOD_CONTROLLER_PORT": str(a["controller_port"]),
        })
        if host in ("localhost", "127.0.0.1"):
            self._workers[wid] = subprocess.Popen(self.command, env=env)
        else:
            import shlex
            exports = " ".join(f"{k}={shlex.quote(v)}"
                               for k, v in env.items()
                               if k.startswith(("HOROVOD_", "PYTHON", "PATH")))
            cmd = ["ssh", "-o", "StrictHostKeyChecking=no", host,
                   f"cd {shlex.quote(os.getcwd())} && env {exports} " +
                   " ".join(shlex.quote(c) for c in self.command)]
            self._workers[wid] = subprocess.Popen(cmd)

    def _reap_workers(self):
        """Handle worker exits (reference: driver.py:304-321
        _handle_worker_exit)."""
        with self._lock:
            done, failed = [], []
            for wid, p in self._workers.items():
                rc = p.poll()
                if rc is None:
                    continue
                if wid not in self._assignments:
                    done.append(wid)  # terminated stale worker
                elif rc == 0:
                    self._finish(None)  # success: job complete
                    return
                else:
                    failed.append(wid)
            for wid in done:
                del self._workers[wid]
            if failed:
                for wid in failed:
                    host = wid.rsplit(":", 1)[0]
                    self._blacklist[host] = time.time()
                    del self._workers[wid]
                hosts = self._discover()
                if hosts:
                    self._update_assignments(hosts)
                else:
                    self._finish(RuntimeError("elastic: no usable hosts"))

    def _finish(self, error):
        self._result = error
        self._result_event.set()


def _my_addr():
    import socket
    return socket.gethostbyname(socket.gethostname())


def run_elastic(args, command, env):
    """Entry from `hvdrun --host-discovery-script ...`."""
    discovery = HostDiscoveryScript(args.host_discovery_script)
    driver = ElasticDriver(discovery, command, env=env,
                           min_np=args.min_np or 1,
                           max_np=args.max_np)
    driver.start()
    err = driver.wait_for_result()
    driver.stop()
    if err is not None:
        print(f"hvdrun elastic: {err}", file=sys.stderr)
        return 1
    return 0
