"""Ray integration (reference: horovod/ray/runner.py RayExecutor —
coordinator collects host/rank info and sets the slot env per worker actor).

ray is not bundled with the MI355X image; the executor degrades to an
informative ImportError when absent.
"""
import os


def _require_ray():
    try:
        import ray
        return ray
    except ImportError as e:
        raise ImportError("horovod_amd.ray requires the ray package") from e


class RayExecutor:
    """Launch horovod_amd workers as Ray actors (reference:
    ray/runner.py:45-423)."""

    def __init__(self, num_workers=1, use_gpu=False, cpus_per_worker=1,
                 gpus_per_worker=None, settings=None):
        self.num_workers = num_workers
        self.use_gpu = use_gpu
        self.cpus_per_worker = cpus_per_worker
        self.gpus_per_worker = gpus_per_worker if gpus_per_worker is not None \
            else (1 if use_gpu else 0)
        self.workers = []

    def start(self):
        ray = _require_ray()
        from horovod_amd.runner.launch import find_free_port

        @ray.remote(num_cpus=self.cpus_per_worker,
                    num_gpus=self.gpus_per_worker)
        class Worker:
            def node_ip(self):
                import ray.util
                return ray.util.get_node_ip_address()

            def set_env(self, env):
                os.environ.update(env)

            def run(self, fn, args, kwargs):
                return fn(*args, **kwargs)

        self.workers = [Worker.remote() for _ in range(self.num_workers)]
        ips = ray.get([w.node_ip.remote() for w in self.workers])
        controller = ips[0]
        port = find_free_port()
        uniq = list(dict.fromkeys(ips))
        local_counts = {}
        envs = []
        for rank, ip in enumerate(ips):
            li = local_counts.get(ip, 0)
            local_counts[ip] = li + 1
            envs.append({
                "HOROVOD_RANK": str(rank),
                "HOROVOD_SIZE": str(self.num_workers),
                "HOROVOD_LOCAL_RANK": str(li),
                "HOROVOD_CROSS_RANK": str(uniq.index(ip)),
                "HOROVOD_CROSS_SIZE": str(len(uniq)),
                "HOROVOD_CONTROLLER_ADDR": controller,
                "HOROVOD_CONTROLLER_PORT": str(port),
            })
        for rank, env in enumerate(envs):
            env["HOROVOD_LOCAL_SIZE"] = str(local_counts[ips[rank]])
        ray.get([w.set_env.remote(e) for w, e in zip(self.workers, envs)])

    def run(self, fn, args=(), kwargs=None):
        ray = _require_ray()
        kwargs = kwargs or {}
        return ray.get([w.run.remote(fn, args, kwargs) for w in self.workers])

    def execute(self, fn):
        return self.run(lambda: fn(None))

    def shutdown(self):
        ray = _require_ray()
        for w in self.workers:
            ray.kill(w)
        self.workers = []


class RayHostDiscovery:
    """HostDiscovery over Ray's cluster view (reference:
    ray/elastic_v2.py:40-110 RayHostDiscovery): alive nodes -> slots from
    their GPU (or CPU) resources."""

    def __init__(self, use_gpu=False, cpus_per_worker=1, gpus_per_worker=1):
        self.use_gpu = use_gpu
        self.cpus_per_worker = cpus_per_worker
        self.gpus_per_worker = gpus_per_worker

    def find_available_hosts_and_slots(self):
        ray = _require_ray()
        hosts = {}
        for node in ray.nodes():
            if not node.get("Alive"):
                continue
            res = node.get("Resources", {})
            host = node.get("NodeManagerAddress")
            if self.use_gpu:
                slots = int(res.get("GPU", 0) // max(self.gpus_per_worker, 1))
            else:
                slots = int(res.get("CPU", 0) // max(self.cpus_per_worker, 1))
            if host and slots > 0:
                hosts[host] = slots
        return hosts


class ElasticRayExecutor:
    """Elastic training on a Ray cluster (reference: ray/elastic_v2.py
    ElasticAdapter): the elastic driver follows Ray's node membership via
    RayHostDiscovery; workers are launched onto the discovered hosts with
    the standard elastic rendezvous env (ssh/local spawn — every Ray node
    of a typical cluster is ssh-reachable, and single-node clusters spawn
    locally)."""

    def __init__(self, use_gpu=False, cpus_per_worker=1, gpus_per_worker=1,
                 min_np=1, max_np=None, reset_limit=None,
                 override_discovery=None):
        self.discovery = override_discovery or RayHostDiscovery(
            use_gpu, cpus_per_worker, gpus_per_worker)
        self.min_np = min_np
        self.max_np = max_np
        self.reset_limit = reset_limit
        self._driver = None

    def start(self):
        # fail early with a clear error if ray is absent (unless a custom
        # discovery was injected, e.g. in tests or hybrid clusters)
        if isinstance(self.discovery, RayHostDiscovery):
            _require_ray()

    def run(self, fn, args=(), kwargs=None):
        """Run `fn` (an hvd.elastic-decorated callable or any function using
        the elastic API) across the cluster until completion."""
        import pickle
        import sys
        import tempfile

        import cloudpickle

        from horovod_amd.runner.elastic_driver import ElasticDriver

        payload = cloudpickle.dumps((fn, tuple(args), dict(kwargs or {})))
        fd, fn_path = tempfile.mkstemp(suffix=".pkl", dir=os.getcwd())
        with os.fdopen(fd, "wb") as f:
            f.write(payload)
        out_path = fn_path + ".out"
        command = [sys.executable, "-m", "horovod_amd.spark._elastic_task",
                   fn_path, out_path]
        self._driver = ElasticDriver(self.discovery, command,
                                     min_np=self.min_np, max_np=self.max_np,
                                     reset_limit=self.reset_limit)
        self._driver.start()
        err = self._driver.wait_for_result()
        self._driver.stop()
        try:
            if err is not None:
                raise err if isinstance(err, Exception) \
                    else RuntimeError(err)
            with open(out_path, "rb") as f:
                return pickle.load(f)
        finally:
            for p in (fn_path, out_path):
                try:
                    os.unlink(p)
                except OSError:
                    pass

    def shutdown(self):
        if self._driver is not None:
            self._driver.stop()
            self._driver = None
