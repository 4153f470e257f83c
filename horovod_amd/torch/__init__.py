"""horovod_amd.torch — the `hvd` API surface for PyTorch on MI355X.

Usage mirrors the reference (horovod/torch/__init__.py):

    import horovod_amd.torch as hvd
    hvd.init()
    opt = hvd.DistributedOptimizer(opt, named_parameters=model.named_parameters())
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
"""
from horovod_amd.torch.compression import (Compression,  # noqa: F401
                                            Compressor, FP16Compressor,
                                            BF16Compressor, NoneCompressor)
from horovod_amd.torch.functions import (allgather_object,  # noqa: F401
                                         broadcast_object,
                                         broadcast_optimizer_state,
                                         broadcast_parameters)
from horovod_amd.torch.mpi_ops import (  # noqa: F401
    Adasum, Average, Max, Min, Product, Sum,
    allgather, allgather_async, allreduce, allreduce_, allreduce_async,
    allreduce_async_, alltoall, alltoall_async, barrier, broadcast,
    broadcast_, broadcast_async, broadcast_async_, ccl_built, cross_rank,
    cross_size, cuda_built, ddl_built, gloo_built, gloo_enabled,
    grouped_allgather, grouped_allgather_async, grouped_allreduce,
    grouped_allreduce_, grouped_allreduce_async, grouped_allreduce_async_,
    grouped_reducescatter, grouped_reducescatter_async,
    handle_average_backwards_compatibility, init, is_homogeneous,
    is_initialized, join, local_rank, local_size, mpi_built, mpi_enabled,
    mpi_threads_supported, nccl_built, poll, rank, reducescatter,
    process_set_included, reducescatter_async, rocm_built, shutdown, size,
    sparse_allreduce_async, start_timeline, stop_timeline, synchronize, wait)
from horovod_amd.torch.mpi_ops import (  # noqa: F401  (autograd classes)
    HorovodAllgather, HorovodAllreduce, HorovodAlltoall, HorovodBroadcast,
    HorovodGroupedAllreduce, HorovodReducescatter)
from horovod_amd.torch.mpi_ops import (add_process_set,  # noqa: F401
                                       remove_process_set)
from horovod_amd.common.process_sets import (ProcessSet,  # noqa: F401
                                             global_process_set)
from horovod_amd.common.exceptions import HorovodInternalError  # noqa: F401
from horovod_amd.torch.lr_scheduler import WarmupScheduler  # noqa: F401
from horovod_amd.torch.metrics import MetricAverager, avg_metrics  # noqa: F401
from horovod_amd.torch.optimizer import DistributedOptimizer  # noqa: F401
from horovod_amd.torch.sync_batch_norm import SyncBatchNorm  # noqa: F401

from horovod_amd.torch import elastic  # noqa: F401  (hvd.elastic.*)

__version__ = "0.1.0"
