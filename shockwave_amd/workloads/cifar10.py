"""ResNet-18 / CIFAR-10 workload — the flagship job family.

Rebuild of the reference's three cifar10 mains
(workloads/pytorch/image_classification/cifar10/main.py and its accordion
/gns twins) as one entry with an adaptation-mode flag.  Training math:
SGD momentum 0.9, weight decay 5e-4, cross-entropy, per-epoch loop under
the LeaseIterator; Accordion/GNS via the shared adaptation library backed
by the CDNA4 kernels.
"""

from __future__ import annotations

import argparse
import math
import os
import sys

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, distributed

from ..adapt import AccordionDetector, GNSEstimator, hardcoded_critical_regime
from ..core import datasets as ds_tables
from ..data.synthetic import SyntheticImages
from ..models import resnet18_cifar
from ..ops.optim import FusedSGD
from . import common


def parse_args(argv=None):
    p = argparse.ArgumentParser(description="ResNet-18/CIFAR-10 under lease")
    p.add_argument("--data_dir", type=str, default=None)
    p.add_argument("--batch_size", type=int, default=128)
    p.add_argument("--lr", type=float, default=0.1)
    p.add_argument("--momentum", type=float, default=0.9)
    p.add_argument("--weight_decay", type=float, default=5e-4)
    common.add_scheduler_args(p, "--num_steps")
    return p.parse_args(argv)


def linear_learning_rate(base_lr, batch_size, base_bs=128):
    """Linear LR scaling with batch size (reference accordion
    main.py:383)."""
    return base_lr * batch_size / base_bs


def main(argv=None, mode=None, client=None, max_steps_override=None):
    args = parse_args(argv)
    mode = args.mode or mode or os.environ.get("SWQ_MODE", "static")
    device = common.init_device_and_distributed(args)

    dataset = SyntheticImages(
        ds_tables.dataset_len("CIFAR-10"), image_size=32, num_classes=10
    )
    sampler = None
    if args.world_size > 1:
        sampler = distributed.DistributedSampler(
            dataset, num_replicas=args.world_size, rank=args.rank
        )
    loader = DataLoader(
        dataset,
        batch_size=args.batch_size,
        shuffle=(sampler is None),
        sampler=sampler,
        num_workers=0,
        drop_last=True,
    )

    model = resnet18_cifar().to(device)
    model_train = common.wrap_distributed(model, args)
    criterion = nn.CrossEntropyLoss().to(device)
    lr = linear_learning_rate(args.lr, args.batch_size)
    optimizer = FusedSGD(
        model_train.parameters(), lr=lr, momentum=args.momentum,
        weight_decay=args.weight_decay,
    )

    trainloader, lease_it = common.make_lease_iterator(
        loader, args, synthetic_data=True, client=client
    )

    # --- checkpoint restore -------------------------------------------------
    steps_per_epoch = len(loader)
    start_epoch, cumulative_steps = 0, 0
    accordion = gns = None
    original_bs = args.batch_size
    if mode == "accordion":
        accordion = AccordionDetector(model)
    elif mode == "gns":
        window = args.world_size if args.world_size > 1 else 2
        gns = GNSEstimator(model, args.batch_size, window=window)

    ckpt = None
    if lease_it is not None:
        ckpt = lease_it.load_checkpoint()
    if ckpt:
        model.load_state_dict(ckpt["model"])
        try:
            optimizer.load_state_dict(ckpt["optimizer"])
        except (ValueError, KeyError):
            pass  # batch-size rescale can change optimizer hyperparams
        start_epoch = ckpt.get("epoch", 0)
        cumulative_steps = ckpt.get("cumulative_steps", 0)
        original_bs = ckpt.get("original_bs", original_bs)
        if accordion is not None and "accordion" in ckpt:
            accordion.load_state_dict(ckpt["accordion"])
        if gns is not None and "gns" in ckpt:
            gns.load_state_dict(ckpt["gns"])

    target_steps = max_steps_override or args.num_steps or (steps_per_epoch * 5)
    reporter = common.ThroughputReporter(
        args.throughput_estimation_interval, args.rank
    )

    def save(epoch):
        if args.rank != 0 or lease_it is None:
            return
        state = {
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict(),
            "epoch": epoch,
            "cumulative_steps": cumulative_steps,
            "original_bs": original_bs,
        }
        if accordion is not None:
            state["accordion"] = accordion.state_dict()
        if gns is not None:
            state["gns"] = gns.state_dict()
        lease_it.save_checkpoint(state)

    # --- training loop ------------------------------------------------------
    model_train.train()
    epoch = start_epoch
    done = False
    while not done and cumulative_steps < target_steps:
        hit_target = False
        for inputs, targets in trainloader:
            inputs = inputs.to(device, non_blocking=True)
            targets = targets.to(device, non_blocking=True)
            common.zero_grads(model_train)
            outputs = model_train(inputs)
            loss = criterion(outputs, targets)
            loss.backward()
            common.finish_sync(model_train)
            if accordion is not None:
                accordion.on_step()
            if gns is not None:
                gns.on_step()
            optimizer.step()
            cumulative_steps += 1
            reporter.step()
            if cumulative_steps >= target_steps:
                hit_target = True
                break
        if hit_target or (lease_it is not None and lease_it.done):
            break
        # epoch boundary housekeeping (lease not expired)
        if accordion is not None:
            accordion.on_epoch(epoch)
            in_cr = hardcoded_critical_regime(
                "ResNet-18", original_bs, epoch + 1
            )
            max_bs = ds_tables.max_batch_size("ResNet-18")
            if (
                not in_cr
                and args.batch_size == original_bs
                and args.batch_size != max_bs
                and lease_it is not None
            ):
                lease_it.update_resource_requirement(True, False)
                done = True
            elif (
                in_cr
                and args.batch_size != original_bs
                and lease_it is not None
            ):
                lease_it.update_resource_requirement(False, True)
                done = True
        if gns is not None:
            gns.on_epoch(epoch)
            max_bs = ds_tables.max_batch_size("ResNet-18")
            if (
                gns.should_double(epoch)
                and args.batch_size < max_bs
                and lease_it is not None
            ):
                lease_it.update_resource_requirement(True, False)
                done = True
        epoch += 1

    if lease_it is not None:
        save(epoch)
        if cumulative_steps >= target_steps and not lease_it.done:
            lease_it.complete()
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()
    return cumulative_steps


if __name__ == "__main__":
    sys.exit(0 if main() is not None else 1)
