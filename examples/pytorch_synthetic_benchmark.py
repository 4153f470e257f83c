"""Synthetic benchmark matching the reference's
examples/pytorch/pytorch_synthetic_benchmark.py flag surface (13-35):

    hvdrun -np 8 python examples/pytorch_synthetic_benchmark.py \
        --model resnet50 --batch-size 64 [--fp16-allreduce] [--use-adasum]
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse
import timeit

import numpy as np
import torch

import horovod_amd.torch as hvd

parser = argparse.ArgumentParser(description="PyTorch Synthetic Benchmark",
                                 formatter_class=argparse.ArgumentDefaultsHelpFormatter)
parser.add_argument("--model", type=str, default="resnet50")
parser.add_argument("--batch-size", type=int, default=32)
parser.add_argument("--num-warmup-batches", type=int, default=10)
parser.add_argument("--num-batches-per-iter", type=int, default=10)
parser.add_argument("--num-iters", type=int, default=10)
parser.add_argument("--fp16-allreduce", action="store_true",
                    help="use fp16 compression during allreduce")
parser.add_argument("--bf16-allreduce", action="store_true",
                    help="use bf16 compression during allreduce")
parser.add_argument("--use-adasum", action="store_true",
                    help="use the Adasum reduction")
parser.add_argument("--no-cuda", action="store_true")
args = parser.parse_args()

args.cuda = not args.no_cuda and torch.cuda.is_available()

hvd.init()
if args.cuda:
    torch.cuda.set_device(hvd.local_rank())
    torch.backends.cudnn.benchmark = True

from horovod_amd.models import resnet50, resnet101, resnet152  # noqa: E402

model = {"resnet50": resnet50, "resnet101": resnet101,
         "resnet152": resnet152}[args.model]()
if args.cuda:
    model.cuda()
    model = model.to(memory_format=torch.channels_last)

optimizer = torch.optim.SGD(model.parameters(), lr=0.01 * hvd.size())
compression = (hvd.Compression.fp16 if args.fp16_allreduce else
               hvd.Compression.bf16 if args.bf16_allreduce else
               hvd.Compression.none)
optimizer = hvd.DistributedOptimizer(
    optimizer, named_parameters=model.named_parameters(),
    compression=compression,
    op=hvd.Adasum if args.use_adasum else hvd.Average)

hvd.broadcast_parameters(model.state_dict(), root_rank=0)
hvd.broadcast_optimizer_state(optimizer, root_rank=0)

if args.cuda:
    data = torch.randn(args.batch_size, 3, 224, 224).cuda().to(
        memory_format=torch.channels_last)
    target = torch.randint(0, 1000, (args.batch_size,)).cuda()
else:
    data = torch.randn(args.batch_size, 3, 64, 64)
    target = torch.randint(0, 1000, (args.batch_size,))


def benchmark_step():
    optimizer.zero_grad()
    with torch.autocast("cuda", dtype=torch.bfloat16, enabled=args.cuda):
        output = model(data)
        loss = torch.nn.functional.cross_entropy(output, target)
    loss.backward()
    optimizer.step()


def log(s):
    if hvd.rank() == 0:
        print(s, flush=True)


log(f"Model: {args.model}")
log(f"Batch size: {args.batch_size}")
log(f"Number of GPUs: {hvd.size()}" if args.cuda else
    f"Number of CPU workers: {hvd.size()}")

log("Running warmup...")
timeit.timeit(benchmark_step, number=args.num_warmup_batches)

log("Running benchmark...")
img_secs = []
for x in range(args.num_iters):
    time = timeit.timeit(benchmark_step, number=args.num_batches_per_iter)
    img_sec = args.batch_size * args.num_batches_per_iter / time
    log(f"Iter #{x}: {img_sec:.1f} img/sec per worker")
    img_secs.append(img_sec)

img_sec_mean = np.mean(img_secs)
img_sec_conf = 1.96 * np.std(img_secs)
log(f"Img/sec per worker: {img_sec_mean:.1f} +-{img_sec_conf:.1f}")
log(f"Total img/sec on {hvd.size()} worker(s): "
    f"{hvd.size() * img_sec_mean:.1f} +-{hvd.size() * img_sec_conf:.1f}")
