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


def run_elastic(*a, **kw):
    raise NotImplementedError(
        "elastic Spark jobs: use hvdrun --host-discovery-script with a "
        "script that queries the Spark master for executor hosts")
