#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 synthetic-ImageNet training throughput.

Matches BASELINE.json's metric ("images/sec/GPU + scaling efficiency,
ResNet-50 synthetic at 1/2/4/8 MI355X") and the reference's benchmark method
(examples/pytorch/pytorch_synthetic_benchmark.py: synthetic data, batch
64/GPU, data-parallel hvd.DistributedOptimizer).

Launch (driver contract):
  python bench.py --gpus N --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N ...

One JSON line on rank 0; `value` is the WHOLE-JOB total images/sec.
"""
import argparse
import json
import os
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch-size", type=int, default=64, help="per-GPU batch")
    p.add_argument("--model", default="resnet50",
                   choices=["resnet50", "resnet101", "resnet152",
                            "bert-large", "bert-base"])
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--image-size", type=int, default=224)
    p.add_argument("--compression", default="none",
                   choices=["none", "fp16", "bf16"])
    p.add_argument("--use-adasum", action="store_true")
    p.add_argument("--no-bf16", action="store_true",
                   help="disable bf16 autocast (fp32 compute)")
    p.add_argument("--optimizer", default="sgd",
                   choices=["sgd", "adamw"],
                   help="adamw uses the fused CDNA4 AdamW kernel")
    p.add_argument("--fused-sgd", dest="fused_sgd", action="store_true",
                   default=True,
                   help="use the CDNA4 fused SGD step kernel (default)")
    p.add_argument("--no-fused-sgd", dest="fused_sgd", action="store_false")
    p.add_argument("--fused-bn", action="store_true",
                   help="use the fused CDNA4 BatchNorm(+Add)+ReLU kernels "
                        "(experimental: numerics-verified, currently slower "
                        "than MIOpen's BN at batch 64)")
    p.add_argument("--persistent-grads", action="store_true",
                   help="zero_grad(set_to_none=False): keep gradient buffers "
                        "allocated across steps")
    p.add_argument("--hipgraph", action="store_true",
                   help="capture forward+backward in a hipGraph; allreduce + "
                        "optimizer run after replay (trades overlap for "
                        "launch overhead)")
    return p.parse_args()


def main():
    args = parse_args()
    import horovod_amd.torch as hvd
    hvd.init()

    cuda = torch.cuda.is_available()
    if cuda:
        torch.cuda.set_device(hvd.local_rank())
        torch.backends.cudnn.benchmark = True
        device = torch.device("cuda", hvd.local_rank())
    else:
        device = torch.device("cpu")

    from horovod_amd.models import (bert_base, bert_large, resnet50,
                                    resnet101, resnet152)
    is_bert = args.model.startswith("bert")
    model_fn = {"resnet50": resnet50, "resnet101": resnet101,
                "resnet152": resnet152, "bert-large": bert_large,
                "bert-base": bert_base}[args.model]
    torch.manual_seed(42)
    model = (model_fn(fused_bn=True) if (args.fused_bn and not is_bert)
             else model_fn()).to(device)
    if cuda and not is_bert:
        model = model.to(memory_format=torch.channels_last)

    if is_bert:
        batch = args.batch_size if cuda else 2
        seq = args.seq_len if cuda else 32
        image_size = None
        data = torch.randint(0, 30522, (batch, seq), device=device)
        target = torch.randint(0, 30522, (batch, seq), device=device)
    else:
        batch = args.batch_size if cuda else 8
        image_size = args.image_size if cuda else 64
        data = torch.randn(batch, 3, image_size, image_size, device=device)
        if cuda:
            data = data.to(memory_format=torch.channels_last)
        target = torch.randint(0, 1000, (batch,), device=device)

    from horovod_amd.torch.compression import Compression
    compression = {"none": Compression.none, "fp16": Compression.fp16,
                   "bf16": Compression.bf16}[args.compression]

    if args.optimizer == "adamw":
        from horovod_amd.ops import FusedAdamW
        opt = FusedAdamW(model.parameters(), lr=1e-4 * hvd.size(),
                         weight_decay=0.01)
    elif args.fused_sgd:
        from horovod_amd.ops import FusedSGD
        opt = FusedSGD(model.parameters(), lr=0.0125 * hvd.size(),
                       momentum=0.9, weight_decay=5e-5)
    else:
        opt = torch.optim.SGD(model.parameters(), lr=0.0125 * hvd.size(),
                              momentum=0.9, weight_decay=5e-5)
    loss_fn = torch.nn.CrossEntropyLoss()
    use_bf16 = cuda and not args.no_bf16

    if args.hipgraph and cuda:
        # hipGraph mode: fwd+bwd captured once and replayed (kills per-kernel
        # launch overhead on the 2000-dispatch ResNet step); the gradient
        # allreduce runs after replay as ONE grouped op, then the (fused) SGD
        # step.  Trades backward/comm overlap for CPU-side launch cost.
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        params = [p for p in model.parameters() if p.requires_grad]
        for p in params:
            p.grad = torch.zeros_like(p)
        wire = compression.wire_dtype(torch.float32)

        def fwd_bwd():
            with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16):
                loss = loss_fn(model(data), target)
            loss.backward()
            return loss

        # warm up allocator state on a side stream, then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                for p in params:
                    p.grad.zero_()
                fwd_bwd()
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        for p in params:
            p.grad.zero_()
        with torch.cuda.graph(graph):
            fwd_bwd()
        grads = [p.grad for p in params]

        from horovod_amd.torch import mpi_ops as _ops

        def step():
            graph.replay()
            # grouped allreduce runs even at n=1 so the timed region always
            # includes the full gradient pipeline (pack -> RCCL -> unpack)
            h = _ops._grouped_allreduce_impl(
                grads, grads, None, "hipgraph_grads", hvd.Average, 1.0,
                1.0, hvd.global_process_set, wire_dtype=wire)
            hvd.synchronize(h)
            opt.step()
            for g in grads:
                g.zero_()
    else:
        opt = hvd.DistributedOptimizer(
            opt, named_parameters=model.named_parameters(),
            compression=compression,
            op=hvd.Adasum if args.use_adasum else hvd.Average)
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        hvd.broadcast_optimizer_state(opt, root_rank=0)

        def step():
            opt.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16, enabled=use_bf16):
                if is_bert:
                    mlm, _ = model(data)
                    loss = loss_fn(mlm.flatten(0, 1), target.flatten())
                else:
                    loss = loss_fn(model(data), target)
            loss.backward()
            opt.step()

    for _ in range(args.warmup):
        step()

    hvd.barrier()
    if cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if cuda:
        torch.cuda.synchronize()
    hvd.barrier()
    elapsed = time.perf_counter() - t0
    # MAX over ranks
    elapsed_t = torch.tensor([elapsed], dtype=torch.float64)
    elapsed = float(hvd.allreduce(elapsed_t, op=hvd.Max,
                                  name="bench_elapsed").item())

    n = hvd.size()
    total_images = n * batch * args.steps
    value = total_images / elapsed
    if hvd.rank() == 0:
        result = {
            "metric": (f"sequences/sec, {args.model} synthetic (total)" if is_bert
                       else "images/sec, ResNet-50 synthetic (total over all GPUs)"),
            "value": round(value, 2),
            "unit": "sequences/sec" if is_bert else "images/sec",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_bf16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": n * batch,
                "image_size": image_size,
                "seq_len": args.seq_len if is_bert else None,
                "parallelism": f"dp{n}",
                "compression": args.compression,
                "reduction": "adasum" if args.use_adasum else "average",
                "optimizer": args.optimizer,
                "fused_sgd": args.fused_sgd,
                "fused_bn": args.fused_bn,
                "hipgraph": args.hipgraph,
            },
        }
        print(json.dumps(result), flush=True)
    hvd.shutdown()


if __name__ == "__main__":
    main()
