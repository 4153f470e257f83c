"""Exercise every custom CDNA4 kernel on one GPU (for rocprofv3 evidence and
bandwidth numbers): batched pack/unpack (+ wire conversion), adasum
dot/scaled-add, fused SGD."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import horovod_amd.torch as hvd  # noqa: E402
from horovod_amd import _core  # noqa: E402
from horovod_amd.ops import FusedSGD  # noqa: E402
from horovod_amd.torch.compression import Compression  # noqa: E402

hvd.init()
torch.cuda.set_device(0)
MB = 1024 * 1024


def timed(label, fn, iters=10, bytes_moved=None):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    bw = f", {bytes_moved / dt / 1e9:8.1f} GB/s" if bytes_moved else ""
    print(f"{label:44s} {dt * 1e3:8.3f} ms{bw}", flush=True)


# 1) fused pack + RCCL + unpack (64 x 1MB fp32 tensors = 64 MB bucket)
ts = [torch.randn(MB // 4, device="cuda") for _ in range(64)]
timed("grouped_allreduce 64x1MB fp32 (pack+unpack)",
      lambda: hvd.grouped_allreduce(ts, average=False, name="kp1"),
      bytes_moved=4 * 64 * MB)

# 2) with bf16 wire conversion fused into pack
timed("grouped_allreduce 64x1MB fp32->bf16 wire",
      lambda: hvd.grouped_allreduce(ts, average=False, name="kp2",
                                    prescale_factor=1.0),
      bytes_moved=4 * 64 * MB)
out = hvd.allreduce(ts[0], average=False, compression=Compression.bf16,
                    name="kp2b")

# 3) adasum combine kernels, 64 MB pair
a = [torch.randn(16 * MB, device="cuda") for _ in range(4)]
b = [torch.randn(16 * MB, device="cuda") for _ in range(4)]
timed("adasum dots+scaledadd 4x64MB fp32",
      lambda: _core.adasum_combine_(a, b), bytes_moved=3 * 4 * 64 * MB)

# 4) fused SGD, ResNet-50-shaped params
params = [torch.randn(n, device="cuda") for n in
          [25_557_032 // 160] * 160]
for p in params:
    p.grad = torch.randn_like(p)
opt = FusedSGD(params, lr=0.01, momentum=0.9, weight_decay=1e-4)
timed("fused SGD step, 160 params (~25M elems)", opt.step,
      bytes_moved=5 * 4 * sum(p.numel() for p in params))

# 5) fused AdamW, BERT-shaped params (p,g,m,v read + p,m,v write = 7 passes)
from horovod_amd.ops import FusedAdamW
aparams = [torch.randn(n, device="cuda") for n in [340_000_000 // 400] * 64]
for p in aparams:
    p.grad = torch.randn_like(p)
aopt = FusedAdamW(aparams, lr=1e-4)
timed("fused AdamW step, 64 params (~54M elems)", aopt.step,
      bytes_moved=7 * 4 * sum(p.numel() for p in aparams))
hvd.shutdown()
