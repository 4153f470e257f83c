"""ImageNet ResNet-50 training with horovod_amd.

The MI355X-native counterpart of the reference's
examples/pytorch/pytorch_imagenet_resnet50.py: one process per GPU
(launch with `bin/hvdrun -np 8 python examples/pytorch_imagenet_resnet50.py
--train-dir ...`), DistributedOptimizer over RCCL/xGMI, bf16 autocast,
channels_last, fused SGD, checkpoint save/resume on rank 0 with
broadcast_parameters/broadcast_optimizer_state.

With --synthetic (default when no --train-dir is given) it trains on
random data of ImageNet shape, so it runs in this offline image.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402
import horovod_amd.torch as hvd  # noqa: E402
from horovod_amd.models import resnet50  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--train-dir", default=None,
                   help="ImageFolder root; omit for synthetic data")
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--steps-per-epoch", type=int, default=50,
                   help="synthetic mode: steps per epoch")
    p.add_argument("--base-lr", type=float, default=0.0125,
                   help="per-GPU lr (scaled by world size, reference-style)")
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--wd", type=float, default=5e-5)
    p.add_argument("--warmup-epochs", type=float, default=5)
    p.add_argument("--checkpoint-format",
                   default="checkpoint-{epoch}.pt")
    p.add_argument("--fp16-allreduce", action="store_true",
                   help="fp16 wire compression for gradients")
    p.add_argument("--use-adasum", action="store_true")
    p.add_argument("--no-bf16", action="store_true")
    return p.parse_args()


def main():
    args = parse_args()
    hvd.init()
    cuda = torch.cuda.is_available()
    if cuda:
        torch.cuda.set_device(hvd.local_rank())
    device = torch.device("cuda", hvd.local_rank()) if cuda \
        else torch.device("cpu")
    torch.manual_seed(42)

    model = resnet50().to(device)
    if cuda:
        model = model.to(memory_format=torch.channels_last)

    lr_scaler = hvd.size() if not args.use_adasum else 1
    if cuda and args.use_adasum:
        lr_scaler = hvd.local_size()
    from horovod_amd.ops import FusedSGD
    base_opt = (FusedSGD if cuda else torch.optim.SGD)(
        model.parameters(), lr=args.base_lr * lr_scaler,
        momentum=args.momentum, weight_decay=args.wd)
    compression = (hvd.Compression.fp16 if args.fp16_allreduce
                   else hvd.Compression.none)
    opt = hvd.DistributedOptimizer(
        base_opt, named_parameters=model.named_parameters(),
        compression=compression,
        op=hvd.Adasum if args.use_adasum else hvd.Average)

    # resume from the latest checkpoint on rank 0, then broadcast
    start_epoch = 0
    for e in range(args.epochs, 0, -1):
        path = args.checkpoint_format.format(epoch=e - 1)
        if hvd.rank() == 0 and os.path.exists(path):
            ck = torch.load(path, map_location=device, weights_only=True)
            model.load_state_dict(ck["model"])
            opt.load_state_dict(ck["optimizer"])
            start_epoch = e
            break
    start_epoch = hvd.broadcast_object(start_epoch, root_rank=0,
                                       name="start_epoch")
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    hvd.broadcast_optimizer_state(opt, root_rank=0, model=model)

    if args.train_dir:
        from torch.utils.data import DataLoader, DistributedSampler
        import torchvision  # noqa: F401  (not in this image; real clusters)
        dataset = torchvision.datasets.ImageFolder(args.train_dir)
        sampler = DistributedSampler(dataset, num_replicas=hvd.size(),
                                     rank=hvd.rank())
        loader = DataLoader(dataset, batch_size=args.batch_size,
                            sampler=sampler, num_workers=4)
        steps = len(loader)
    else:
        loader = None
        steps = args.steps_per_epoch

    use_bf16 = cuda and not args.no_bf16

    def lr_at(epoch, step):
        # linear warmup then stepwise decay (reference adjust_learning_rate)
        progress = epoch + step / steps
        if progress < args.warmup_epochs:
            factor = (progress / args.warmup_epochs) * (lr_scaler - 1) + 1
            factor /= lr_scaler
        else:
            factor = 10 ** -sum(progress >= m for m in (30, 60, 80))
        return args.base_lr * lr_scaler * factor

    for epoch in range(start_epoch, args.epochs):
        model.train()
        t0 = time.time()
        seen = 0
        it = iter(loader) if loader else None
        for step in range(steps):
            if it is not None:
                data, target = next(it)
                data, target = data.to(device), target.to(device)
            else:
                data = torch.randn(args.batch_size, 3, 224, 224,
                                   device=device)
                target = torch.randint(0, 1000, (args.batch_size,),
                                       device=device)
            if cuda:
                data = data.to(memory_format=torch.channels_last)
            for g in opt.param_groups:
                g["lr"] = lr_at(epoch, step)
            opt.zero_grad()
            with torch.autocast("cuda", dtype=torch.bfloat16,
                                enabled=use_bf16):
                loss = F.cross_entropy(model(data), target)
            loss.backward()
            opt.step()
            seen += args.batch_size
        if cuda:
            torch.cuda.synchronize()
        rate = hvd.allreduce(
            torch.tensor([seen / (time.time() - t0)]),
            average=False, name="rate")
        if hvd.rank() == 0:
            print(f"epoch {epoch}: {rate.item():.1f} img/s total, "
                  f"loss {loss.item():.3f}", flush=True)
            torch.save({"model": model.state_dict(),
                        "optimizer": opt.state_dict()},
                       args.checkpoint_format.format(epoch=epoch))


if __name__ == "__main__":
    main()
