"""Fusion pack/unpack kernel bandwidth probe (single GPU).

Times the full fused-allreduce pipeline on big buckets at n=1 (pack ->
ncclAllReduce(self) -> unpack) and the raw batched-copy kernel via a
grouped allreduce of many tensors.  Run under rocprofv3 for per-kernel
numbers; committed summaries live in profiles/.
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import time

import torch

import horovod_amd.torch as hvd

hvd.init()
torch.cuda.set_device(0)

MB = 1024 * 1024
for n_tensors, total_mb in ((8, 256), (64, 64), (160, 64)):
    per = total_mb * MB // 4 // n_tensors
    ts = [torch.randn(per, device="cuda") for _ in range(n_tensors)]
    for _ in range(3):
        hvd.grouped_allreduce(ts, average=False, name=f"warm{n_tensors}{total_mb}")
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 10
    for i in range(iters):
        hvd.grouped_allreduce(ts, average=False, name=f"p{n_tensors}{total_mb}")
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    # pipeline moves >= 4x the payload through HBM (pack r+w, unpack r+w),
    # plus RCCL's own copy at n=1
    gbps = 4 * total_mb / 1024 / dt
    print(f"{n_tensors:4d} tensors x {total_mb:4d} MB: {dt*1e3:7.2f} ms/op, "
          f"pack+unpack traffic >= {gbps:7.1f} GB/s", flush=True)
hvd.shutdown()
