"""Extended randomized stress (not collected by pytest): mixed collectives
with random sizes/ops across seeds and world sizes.

    for seed in 7 99; do for np in 2 3 4; do
      FUZZ_SEED=$seed python -m horovod_amd.runner.launch -np $np \
          python tests/stress_fuzz.py; done; done
"""
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402
import horovod_amd.torch as hvd  # noqa: E402
from horovod_amd.torch.compression import Compression  # noqa: E402

hvd.init()
rank, size = hvd.rank(), hvd.size()
rng = random.Random(int(os.environ.get("FUZZ_SEED", 7)))
ITERS = int(os.environ.get("FUZZ_ITERS", 150))

for i in range(ITERS):
    kind = rng.choice(["allreduce", "broadcast", "reducescatter", "barrier",
                       "grouped", "alltoall", "compressed", "allgather"])
    n = rng.randint(1, 2000)
    base = torch.arange(n).float()
    mine = base * (rank + 1)
    if kind == "allreduce":
        op = rng.choice([hvd.Sum, hvd.Min, hvd.Max])
        out = hvd.allreduce(mine, op=op, name=f"f{i}")
        exp = {hvd.Sum: base * (size * (size + 1) / 2),
               hvd.Min: base, hvd.Max: base * size}[op]
        assert torch.allclose(out, exp), (i, kind)
    elif kind == "allgather":
        rows = rng.randint(0, 4) + rank
        out = hvd.allgather(torch.full((rows, 2), float(rank)), name=f"f{i}")
        expected_rows = sum(rng_rows + r for r in range(size)
                            for rng_rows in [rows - rank])
        assert out.shape == (expected_rows, 2), (i, out.shape)
    elif kind == "broadcast":
        root = rng.randint(0, size - 1)
        out = hvd.broadcast(mine, root_rank=root, name=f"f{i}")
        assert torch.allclose(out, base * (root + 1)), (i, kind)
    elif kind == "reducescatter":
        out = hvd.reducescatter(mine, op=hvd.Sum, name=f"f{i}")
        lo = rank * (n // size) + min(rank, n % size)
        rows = n // size + (1 if rank < n % size else 0)
        assert torch.allclose(out, base[lo:lo + rows] *
                              (size * (size + 1) / 2)), (i, kind)
    elif kind == "barrier":
        hvd.barrier()
    elif kind == "alltoall":
        per = rng.randint(1, 5)
        t = torch.arange(per * size).float() + rank * 1000
        out, rs = hvd.alltoall(t, splits=[per] * size, name=f"f{i}")
        expected = torch.cat([torch.arange(per * rank, per * (rank + 1)) +
                              r * 1000 for r in range(size)]).float()
        assert torch.allclose(out, expected), (i, kind)
    elif kind == "compressed":
        out = hvd.allreduce(mine, average=False, name=f"f{i}",
                            compression=Compression.fp16)
        exp = base * (size * (size + 1) / 2)
        assert torch.allclose(out, exp, rtol=2e-2, atol=2e-1), (i, kind)
    else:
        outs = hvd.grouped_allreduce([mine, mine * 2], average=False,
                                     name=f"f{i}")
        assert torch.allclose(outs[1], base * size * (size + 1)), (i, kind)

print("FUZZ_OK", rank, flush=True)
