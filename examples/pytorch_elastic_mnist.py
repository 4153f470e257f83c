"""Elastic MNIST example (reference: examples/elastic/pytorch/
pytorch_mnist_elastic.py):

    hvdrun --host-discovery-script ./discover_hosts.sh \
        --min-np 1 --max-np 8 python examples/pytorch_elastic_mnist.py
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import argparse

import torch
import torch.nn.functional as F
import torch.utils.data

import horovod_amd.torch as hvd
import horovod_amd.torch.elastic as elastic
from horovod_amd.models import MNISTNet
from horovod_amd.torch.elastic import ElasticSampler

parser = argparse.ArgumentParser()
parser.add_argument("--batch-size", type=int, default=64)
parser.add_argument("--epochs", type=int, default=2)
parser.add_argument("--lr", type=float, default=0.01)
parser.add_argument("--samples", type=int, default=1024)
args = parser.parse_args()

hvd.init()
torch.manual_seed(0)
cuda = torch.cuda.is_available()
if cuda:
    torch.cuda.set_device(hvd.local_rank())
device = torch.device("cuda", hvd.local_rank()) if cuda else torch.device("cpu")

images = torch.randn(args.samples, 1, 28, 28)
labels = torch.randint(0, 10, (args.samples,))
dataset = torch.utils.data.TensorDataset(images, labels)
sampler = ElasticSampler(dataset)
loader = torch.utils.data.DataLoader(dataset, batch_size=args.batch_size,
                                     sampler=sampler)

model = MNISTNet().to(device)
optimizer = torch.optim.SGD(model.parameters(), lr=args.lr * hvd.size())
optimizer = hvd.DistributedOptimizer(
    optimizer, named_parameters=model.named_parameters())


@elastic.run
def train(state):
    for state.epoch in range(state.epoch, args.epochs):
        state.sampler.set_epoch(state.epoch)
        for batch_idx, (data, target) in enumerate(loader):
            data, target = data.to(device), target.to(device)
            optimizer.zero_grad()
            loss = F.nll_loss(model(data), target)
            loss.backward()
            optimizer.step()
            state.sampler.record_batch(batch_idx, args.batch_size)
            state.commit()
        if hvd.rank() == 0:
            print(f"epoch {state.epoch} done (size={hvd.size()})", flush=True)


state = elastic.TorchState(model, optimizer, sampler=sampler, epoch=0)
train(state)
if hvd.rank() == 0:
    print("elastic training complete")
