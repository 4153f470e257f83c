"""Per-op latency breakdown at np=1: hvd pipeline vs raw RCCL."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
import horovod_amd.torch as hvd  # noqa: E402

hvd.init()
torch.cuda.set_device(0)


def timed(label, fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print(f"{label:46s} {(time.perf_counter()-t0)/iters*1e3:8.3f} ms/op",
          flush=True)


tiny = torch.ones(16, device="cuda")
big = torch.ones(16 << 20, device="cuda")  # 64 MB
timed("hvd.allreduce tiny (direct RCCL path)",
      lambda: hvd.allreduce(tiny, average=False, name="lt"))
timed("hvd.allreduce 64MB (direct RCCL path)",
      lambda: hvd.allreduce(big, average=False, name="lb"))
timed("hvd.allreduce_async+sync tiny",
      lambda: hvd.synchronize(hvd.allreduce_async(tiny, average=False,
                                                  name="la")))

# raw RCCL reference via torch.distributed (ws=1)
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29712")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")
import torch.distributed as dist  # noqa: E402
dist.init_process_group("nccl", rank=0, world_size=1)
timed("torch.distributed nccl allreduce tiny",
      lambda: dist.all_reduce(tiny))
timed("torch.distributed nccl allreduce 64MB",
      lambda: dist.all_reduce(big))


def tiny_sync():
    dist.all_reduce(tiny)
    torch.cuda.synchronize()


timed("torch.distributed tiny + synchronize", tiny_sync)
hvd.shutdown()


# ---- multi-rank crossover sweep ------------------------------------------
# Run on a multi-GPU box:
#   bin/hvdrun -np 8 python examples/latency_probe.py                 # ring
#   HOROVOD_ONESHOT_ALLREDUCE=1 bin/hvdrun -np 8 python examples/...  # one-shot
# and compare the per-size rows: the crossover where ring beats one-shot is
# the value to hand HOROVOD_ONESHOT_THRESHOLD (the autotuner explores it
# too — docs/autotune.md).
if hvd.size() > 1:
    dev = torch.device("cuda", hvd.local_rank())
    algo = "one-shot<=%s" % os.environ.get("HOROVOD_ONESHOT_THRESHOLD",
                                           "4MiB") \
        if os.environ.get("HOROVOD_ONESHOT_ALLREDUCE") else "rccl-ring"
    if hvd.rank() == 0:
        print(f"\n-- np={hvd.size()} bucket sweep ({algo}) --", flush=True)
    for numel in (1 << 10, 1 << 14, 1 << 17, 1 << 20, 1 << 22):
        t = torch.ones(numel, device=dev)
        label = f"allreduce {numel * 4 // 1024:>8d} KB x np{hvd.size()}"
        for _ in range(5):
            hvd.allreduce(t, average=False, name="sw")
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            hvd.allreduce(t, average=False, name="sw")
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 30 * 1e3
        if hvd.rank() == 0:
            print(f"{label:46s} {dt:8.3f} ms/op", flush=True)
