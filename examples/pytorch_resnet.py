#!/usr/bin/env python3
# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Decentralized ResNet training (reference analog: examples/pytorch_resnet.py).

Trains a ResNet over a data-partitioned dataset with any of the
decentralized optimizers, LR warmup + step decay, checkpointing every epoch
and train/val accuracy reporting. With no network access the default
dataset is a synthetic, deterministically-generated CIFAR10-shaped
classification problem (class-dependent means, so accuracy is a meaningful
signal); pass ``--data-dir`` holding pre-downloaded ``train.pt``/``val.pt``
tensor files to train on real data.

    ./bfrun -np 4 python examples/pytorch_resnet.py --epochs 3
    ./bfrun -np 8 python examples/pytorch_resnet.py --dist-optimizer win_put
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import argparse
import math
import os

import torch
import torch.nn.functional as F

import bluefog_amd as bf
import bluefog_amd.parallel.topology as tu

parser = argparse.ArgumentParser(
    description="PyTorch ResNet Example", formatter_class=argparse.ArgumentDefaultsHelpFormatter
)
parser.add_argument("--model", default="resnet18", choices=["resnet18", "resnet50", "resnet101"])
parser.add_argument("--data-dir", default=None, help="dir with train.pt/val.pt; synthetic if unset")
parser.add_argument("--batch-size", type=int, default=32)
parser.add_argument("--val-batch-size", type=int, default=32)
parser.add_argument("--epochs", type=int, default=3)
parser.add_argument("--base-lr", type=float, default=0.003, help="per-worker lr")
parser.add_argument("--warmup-epochs", type=float, default=1)
parser.add_argument("--momentum", type=float, default=0.9)
parser.add_argument("--wd", type=float, default=5e-5)
parser.add_argument("--seed", type=int, default=42)
parser.add_argument("--train-samples", type=int, default=2048, help="synthetic train size/worker")
parser.add_argument("--val-samples", type=int, default=512)
parser.add_argument("--image-size", type=int, default=32)
parser.add_argument("--num-classes", type=int, default=10)
parser.add_argument(
    "--dist-optimizer",
    default="neighbor_allreduce",
    choices=[
        "neighbor_allreduce",
        "hierarchical_neighbor_allreduce",
        "allreduce",
        "gradient_allreduce",
        "win_put",
        "pushsum",
    ],
)
parser.add_argument("--atc-style", action="store_true", help="adapt-then-combine variant")
parser.add_argument("--disable-dynamic-topology", action="store_true")
parser.add_argument("--checkpoint-format", default="./checkpoint-{epoch}.pth.tar")
parser.add_argument("--no-checkpoint", action="store_true")
args = parser.parse_args()


def make_synthetic(n, training, device):
    """Class-separable synthetic images: per-class mean patterns + noise."""
    g = torch.Generator().manual_seed(args.seed + (0 if training else 1))
    means = torch.randn(args.num_classes, 3, args.image_size, args.image_size, generator=g)
    gr = torch.Generator().manual_seed(
        args.seed + bf.rank() * 1000 + (17 if training else 31)
    )
    y = torch.randint(0, args.num_classes, (n,), generator=gr)
    x = means[y] + 0.8 * torch.randn(
        n, 3, args.image_size, args.image_size, generator=gr
    )
    return torch.utils.data.TensorDataset(x, y)


def main():
    bf.init()
    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    if device.type == "cuda":
        torch.cuda.set_device(bf.local_rank() % torch.cuda.device_count())
        torch.backends.cudnn.benchmark = True
    torch.manual_seed(args.seed)

    if args.data_dir:
        xtr, ytr = torch.load(os.path.join(args.data_dir, "train.pt"))
        xva, yva = torch.load(os.path.join(args.data_dir, "val.pt"))
        train_ds = torch.utils.data.TensorDataset(xtr, ytr)
        val_ds = torch.utils.data.TensorDataset(xva, yva)
        train_sampler = torch.utils.data.distributed.DistributedSampler(
            train_ds, num_replicas=bf.size(), rank=bf.rank()
        )
        val_sampler = torch.utils.data.distributed.DistributedSampler(
            val_ds, num_replicas=bf.size(), rank=bf.rank()
        )
    else:
        train_ds = make_synthetic(args.train_samples, True, device)
        val_ds = make_synthetic(args.val_samples, False, device)
        train_sampler = val_sampler = None

    train_loader = torch.utils.data.DataLoader(
        train_ds, batch_size=args.batch_size, sampler=train_sampler,
        shuffle=train_sampler is None,
    )
    val_loader = torch.utils.data.DataLoader(
        val_ds, batch_size=args.val_batch_size, sampler=val_sampler
    )

    from bluefog_amd.models import resnet18, resnet50, resnet101

    model = {"resnet18": resnet18, "resnet50": resnet50, "resnet101": resnet101}[
        args.model
    ](num_classes=args.num_classes, zero_init_residual=True).to(device)

    optimizer = torch.optim.SGD(
        model.parameters(),
        lr=args.base_lr * bf.size(),
        momentum=args.momentum,
        weight_decay=args.wd,
    )

    base_cls = (
        bf.DistributedAdaptThenCombineOptimizer
        if args.atc_style
        else bf.DistributedAdaptWithCombineOptimizer
    )
    if args.dist_optimizer == "win_put":
        optimizer = bf.DistributedWinPutOptimizer(optimizer, model=model)
    elif args.dist_optimizer == "pushsum":
        optimizer = bf.DistributedPushSumOptimizer(optimizer, model=model)
    elif args.dist_optimizer == "gradient_allreduce":
        optimizer = bf.DistributedGradientAllreduceOptimizer(optimizer, model=model)
    elif args.dist_optimizer == "allreduce":
        optimizer = base_cls(
            optimizer, model=model, communication_type=bf.CommunicationType.allreduce
        )
    elif args.dist_optimizer == "hierarchical_neighbor_allreduce":
        optimizer = base_cls(
            optimizer,
            model=model,
            communication_type=bf.CommunicationType.hierarchical_neighbor_allreduce,
        )
    else:
        optimizer = base_cls(
            optimizer,
            model=model,
            communication_type=bf.CommunicationType.neighbor_allreduce,
        )

    bf.broadcast_parameters(model.state_dict(), root_rank=0)
    bf.broadcast_optimizer_state(optimizer, root_rank=0)

    dyn_gen = None
    if (
        not args.disable_dynamic_topology
        and bf.size() > 1
        and args.dist_optimizer in ("neighbor_allreduce",)
    ):
        dyn_gen = tu.GetDynamicOnePeerSendRecvRanks(bf.load_topology(), bf.rank())

    def set_dynamic():
        if dyn_gen is None:
            return
        send, recv = next(dyn_gen)
        w = 1.0 / (len(recv) + 1)
        optimizer.self_weight = w
        optimizer.src_weights = {r: w for r in recv}
        optimizer.dst_weights = send
        optimizer.enable_topo_check = False

    steps_per_epoch = max(1, len(train_loader))

    def adjust_lr(epoch, batch_idx):
        if epoch < args.warmup_epochs:
            progress = (batch_idx + 1 + epoch * steps_per_epoch) / (
                args.warmup_epochs * steps_per_epoch
            )
            mult = progress * (bf.size() - 1) / bf.size() + 1.0 / bf.size()
        else:
            mult = 10 ** (-sum(epoch >= e for e in (30, 60, 80)))
        for group in optimizer.param_groups:
            group["lr"] = args.base_lr * bf.size() * mult

    def accuracy(output, target):
        return output.argmax(dim=1).eq(target).float().mean().item()

    for epoch in range(args.epochs):
        model.train()
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)
        tr_loss = tr_acc = seen = 0
        for i, (x, y) in enumerate(train_loader):
            adjust_lr(epoch, i)
            set_dynamic()
            x, y = x.to(device), y.to(device)
            optimizer.zero_grad()
            out = model(x)
            loss = F.cross_entropy(out, y)
            loss.backward()
            optimizer.step()
            tr_loss += loss.item() * y.numel()
            tr_acc += accuracy(out, y) * y.numel()
            seen += y.numel()

        model.eval()
        va_loss = va_acc = vseen = 0
        with torch.no_grad():
            for x, y in val_loader:
                x, y = x.to(device), y.to(device)
                out = model(x)
                va_loss += F.cross_entropy(out, y).item() * y.numel()
                va_acc += accuracy(out, y) * y.numel()
                vseen += y.numel()
        # average the metrics over workers
        stats = torch.tensor(
            [tr_loss / seen, tr_acc / seen, va_loss / vseen, va_acc / vseen]
        )
        stats = bf.allreduce(stats, average=True, name="metrics")
        if bf.rank() == 0:
            print(
                f"epoch {epoch}: train loss {stats[0]:.4f} acc {stats[1]:.3f} | "
                f"val loss {stats[2]:.4f} acc {stats[3]:.3f}",
                flush=True,
            )
            if not args.no_checkpoint:
                torch.save(
                    {"model": model.state_dict(), "epoch": epoch},
                    args.checkpoint_format.format(epoch=epoch),
                )


if __name__ == "__main__":
    main()
