"""Spark integration (reference: horovod/spark — runner.py:200-310 `run`,
estimator API).

Workers are hosted inside Spark barrier tasks; the driver computes the slot
env (rank/size/controller address) from the barrier task infos and each task
runs the user function with horovod_amd initialized, mirroring the
reference's driver/task-service handshake with Spark's own barrier
coordination instead of custom RPC.

pyspark is not bundled with the MI355X image; every entry point degrades to
an informative ImportError when Spark is absent.
"""
import os

from horovod_amd.runner.launch import find_free_port


def _require_pyspark():
    try:
        import pyspark  # noqa: F401
        return pyspark
    except ImportError as e:
        raise ImportError(
            "horovod_amd.spark requires pyspark; install it on your Spark "
            "cluster (the MI355X base image does not bundle it)") from e


def run(fn, args=(), kwargs=None, num_proc=None, use_mpi=None, use_gloo=None,
        extra_env=None, verbose=True):
    """Run `fn` on `num_proc` Spark tasks; returns the list of results by
    rank (reference: spark/runner.py:200-310)."""
    _require_pyspark()
    from pyspark import BarrierTaskContext
    from pyspark.sql import SparkSession

    kwargs = kwargs or {}
    spark = SparkSession.builder.getOrCreate()
    sc = spark.sparkContext
    if num_proc is None:
        num_proc = int(sc.defaultParallelism)

    env_base = dict(extra_env or {})

    def mapper(_):
        ctx = BarrierTaskContext.get()
        infos = ctx.getTaskInfos()
        rank = ctx.partitionId()
        size = len(infos)
        # rank 0's host runs the TCP controller; share its address + a port
        # chosen by rank 0 through the barrier allGather.
        if rank == 0:
            port = find_free_port()
            payload = f"{infos[0].address.split(':')[0]}:{port}"
        else:
            payload = ""
        shared = ctx.allGather(payload)
        addr, port = shared[0].split(":")
        hosts = [i.address.split(":")[0] for i in infos]
        local_rank = sum(1 for r in range(rank) if hosts[r] == hosts[rank])
        local_size = hosts.count(hosts[rank])
        uniq = list(dict.fromkeys(hosts))
        os.environ.update(env_base)
        os.environ.update({
            "HOROVOD_RANK": str(rank),
            "HOROVOD_SIZE": str(size),
            "HOROVOD_LOCAL_RANK": str(local_rank),
            "HOROVOD_LOCAL_SIZE": str(local_size),
            "HOROVOD_CROSS_RANK": str(uniq.index(hosts[rank])),
            "HOROVOD_CROSS_SIZE": str(len(uniq)),
            "HOROVOD_CONTROLLER_ADDR": addr,
            "HOROVOD_CONTROLLER_PORT": port,
        })
        result = fn(*args, **kwargs)
        return [(rank, result)]

    rdd = sc.parallelize(range(num_proc), num_proc).barrier()
    results = rdd.mapPartitions(mapper).collect()
    return [r for _, r in sorted(results)]


class SparkExecutorDiscovery:
    """HostDiscovery backed by the SparkContext's live executor list
    (reference: spark/runner.py run_elastic uses Spark's own view of
    executors to drive elasticity)."""

    def __init__(self, spark_context, default_slots=1):
        self._sc = spark_context
        self.default_slots = default_slots

    def find_available_hosts_and_slots(self):
        # executor memory status keys are "host:port"; the driver itself is
        # excluded so workers only land on executors
        status = self._sc._jsc.sc().getExecutorMemoryStatus()
        keys = list(
            self._sc._jvm.scala.collection.JavaConverters
            .mapAsJavaMapConverter(status).asJava().keySet())
        driver_host = self._sc._conf.get("spark.driver.host", "")
        hosts = {}
        for k in keys:
            host = str(k).rsplit(":", 1)[0]
            if host == driver_host and len(keys) > 1:
                continue
            hosts[host] = hosts.get(host, 0) + self.default_slots
        return hosts


def run_elastic(fn, args=(), kwargs=None, num_proc=None, min_np=1,
                max_np=None, slots_per_host=1, reset_limit=None,
                verbose=True):
    """Elastic Spark jobs (reference: spark/runner.py:312+): the elastic
    driver tracks Spark's live executor hosts, launches workers over ssh
    to those hosts (the usual Spark-cluster topology), and workers run
    `fn` under the elastic rendezvous protocol.  `fn` should use
    hvd.elastic.run internally, exactly as with hvdrun
    --host-discovery-script.  Requires a shared filesystem (the function
    is shipped as a cloudpickle file, like the reference's serialized
    train function)."""
    import sys
    import tempfile

    import cloudpickle

    _require_pyspark()
    from pyspark.sql import SparkSession

    from horovod_amd.runner.elastic_driver import ElasticDriver

    spark = SparkSession.builder.getOrCreate()
    sc = spark.sparkContext
    discovery = SparkExecutorDiscovery(sc, default_slots=slots_per_host)

    payload = cloudpickle.dumps((fn, tuple(args), dict(kwargs or {})))
    fd, fn_path = tempfile.mkstemp(suffix=".pkl", dir=os.getcwd())
    with os.fdopen(fd, "wb") as f:
        f.write(payload)
    out_path = fn_path + ".out"
    command = [sys.executable, "-m", "horovod_amd.spark._elastic_task",
               fn_path, out_path]
    driver = ElasticDriver(discovery, command, min_np=min_np,
                           max_np=max_np or num_proc,
                           reset_limit=reset_limit)
    driver.start()
    err = driver.wait_for_result()
    driver.stop()
    try:
        if err is not None:
            raise err if isinstance(err, Exception) else RuntimeError(err)
        import pickle
        with open(out_path, "rb") as f:
            return pickle.load(f)
    finally:
        for p in (fn_path, out_path):
            try:
                os.unlink(p)
            except OSError:
                pass
