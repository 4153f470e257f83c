"""MNIST training example (reference: examples/pytorch/pytorch_mnist.py).

Runs on synthetic MNIST-shaped data (no network access for datasets):

    hvdrun -np 2 python examples/pytorch_mnist.py --epochs 2
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse

import torch
import torch.nn.functional as F
import torch.utils.data

import horovod_amd.torch as hvd
from horovod_amd.models import MNISTNet

parser = argparse.ArgumentParser()
parser.add_argument("--batch-size", type=int, default=64)
parser.add_argument("--epochs", type=int, default=2)
parser.add_argument("--lr", type=float, default=0.01)
parser.add_argument("--momentum", type=float, default=0.5)
parser.add_argument("--seed", type=int, default=42)
parser.add_argument("--use-adasum", action="store_true")
parser.add_argument("--samples", type=int, default=2048,
                    help="synthetic samples per rank")
args = parser.parse_args()

hvd.init()
torch.manual_seed(args.seed)
cuda = torch.cuda.is_available()
if cuda:
    torch.cuda.set_device(hvd.local_rank())
device = torch.device("cuda", hvd.local_rank()) if cuda else torch.device("cpu")

# synthetic MNIST-shaped dataset, seeded identically then sharded by sampler
images = torch.randn(args.samples, 1, 28, 28)
labels = torch.randint(0, 10, (args.samples,))
dataset = torch.utils.data.TensorDataset(images, labels)
sampler = torch.utils.data.distributed.DistributedSampler(
    dataset, num_replicas=hvd.size(), rank=hvd.rank())
loader = torch.utils.data.DataLoader(dataset, batch_size=args.batch_size,
                                     sampler=sampler)

model = MNISTNet().to(device)
lr_scaler = hvd.size() if not args.use_adasum else 1
optimizer = torch.optim.SGD(model.parameters(), lr=args.lr * lr_scaler,
                            momentum=args.momentum)
optimizer = hvd.DistributedOptimizer(
    optimizer, named_parameters=model.named_parameters(),
    op=hvd.Adasum if args.use_adasum else hvd.Average)

hvd.broadcast_parameters(model.state_dict(), root_rank=0)
hvd.broadcast_optimizer_state(optimizer, root_rank=0)

for epoch in range(args.epochs):
    sampler.set_epoch(epoch)
    model.train()
    for batch_idx, (data, target) in enumerate(loader):
        data, target = data.to(device), target.to(device)
        optimizer.zero_grad()
        loss = F.nll_loss(model(data), target)
        loss.backward()
        optimizer.step()
        if batch_idx % 10 == 0 and hvd.rank() == 0:
            print(f"Epoch {epoch} [{batch_idx}/{len(loader)}] "
                  f"loss={loss.item():.4f}", flush=True)

# average final loss across ranks as a "metric"
final = hvd.allreduce(loss.detach(), name="final_loss")
if hvd.rank() == 0:
    print(f"Final averaged loss: {final.item():.4f}")
