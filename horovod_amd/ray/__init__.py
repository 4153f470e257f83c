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
