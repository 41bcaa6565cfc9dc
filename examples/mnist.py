#!/usr/bin/env python
"""End-to-end convergence example with the dear API (reference parity:
examples/mnist/pytorch_mnist.py — the de-facto integration test that DeAR's
decoupled lazy update converges like SGD).

No network access in this environment, so the dataset is synthetic-MNIST:
10 fixed class templates + Gaussian noise — learnable, deterministic,
train/test split.  Launch single- or multi-process:

    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 2 examples/mnist.py
"""
import argparse
import os
import sys

import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader, TensorDataset
from torch.utils.data.distributed import DistributedSampler

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import dear_pytorch_amd as dear  # noqa: E402
from dear_pytorch_amd.models import MnistNet  # noqa: E402


def synthetic_mnist(n, seed):
    # class templates are FIXED (seed 0) so train/test share the distribution;
    # the per-sample noise uses the split's own seed
    gt = torch.Generator().manual_seed(0)
    templates = torch.randn(10, 1, 28, 28, generator=gt)
    g = torch.Generator().manual_seed(seed)
    labels = torch.randint(0, 10, (n,), generator=g)
    x = templates[labels] + 0.35 * torch.randn(n, 1, 28, 28, generator=g)
    return TensorDataset(x, labels)


def metric_average(val, name):
    t = torch.tensor(float(val))
    return dear.allreduce(t, average=True, name=name).item()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--epochs", type=int, default=3)
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--momentum", type=float, default=0.5)
    p.add_argument("--train-size", type=int, default=8000)
    p.add_argument("--test-size", type=int, default=1000)
    args = p.parse_args()

    dear.init()
    rank, world = dear.rank(), dear.size()
    use_gpu = torch.cuda.is_available()
    device = dear.local_device()
    torch.manual_seed(42)

    train_ds = synthetic_mnist(args.train_size, seed=1)
    test_ds = synthetic_mnist(args.test_size, seed=2)
    train_sampler = DistributedSampler(train_ds, num_replicas=world,
                                       rank=rank) if world > 1 else None
    train_loader = DataLoader(train_ds, batch_size=args.batch_size,
                              sampler=train_sampler,
                              shuffle=train_sampler is None)
    test_loader = DataLoader(test_ds, batch_size=256)

    model = MnistNet().to(device)
    dear.broadcast_parameters(model.state_dict(), root_rank=0)
    optimizer = dear.DistributedOptimizer(
        torch.optim.SGD(model.parameters(), lr=args.lr,
                        momentum=args.momentum),
        model=model)
    dear.broadcast_optimizer_state(optimizer, root_rank=0)

    for epoch in range(args.epochs):
        model.train()
        if train_sampler:
            train_sampler.set_epoch(epoch)
        for bidx, (x, y) in enumerate(train_loader):
            x, y = x.to(device), y.to(device)
            optimizer.zero_grad()
            loss = F.nll_loss(model(x), y)
            loss.backward()
            optimizer.step()
            if bidx % 20 == 0 and rank == 0:
                print(f"epoch {epoch} batch {bidx}: loss {loss.item():.4f}",
                      flush=True)
        # evaluation with cross-rank averaged metrics (dear.allreduce)
        optimizer.synchronize()
        model.eval()
        test_loss, correct, n = 0.0, 0, 0
        with torch.no_grad():
            for x, y in test_loader:
                x, y = x.to(device), y.to(device)
                out = model(x)
                test_loss += F.nll_loss(out, y, reduction="sum").item()
                correct += (out.argmax(1) == y).sum().item()
                n += y.numel()
        test_loss = metric_average(test_loss / n, "avg_loss")
        acc = metric_average(correct / n, "avg_acc")
        if rank == 0:
            print(f"epoch {epoch}: test loss {test_loss:.4f}, "
                  f"accuracy {acc * 100:.2f}%", flush=True)
    dear.shutdown()
    return acc if rank == 0 else None


if __name__ == "__main__":
    main()
