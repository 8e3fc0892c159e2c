"""The seven workload families on the generic lease loop.

CLI flag spellings per family match the reference job-table command
templates (core/job_table.py / reference job_table.py:1-130), so trace
commands dispatch unchanged.
"""

from __future__ import annotations

import argparse

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, distributed

from ..data import synthetic
from ..models import (
    ActorCritic,
    Discriminator,
    GeneratorResNet,
    LSTMLanguageModel,
    RecommendationAutoencoder,
    TranslationTransformer,
    resnet18_cifar,
    resnet50_imagenet,
)
from ..ops.optim import FusedAdam, FusedSGD
from . import common
from .loop import WorkloadSpec, run


def _maybe_sampler(dataset, args):
    if args.world_size > 1:
        return distributed.DistributedSampler(
            dataset, num_replicas=args.world_size, rank=args.rank
        )
    return None


def _loader(dataset, args, bs):
    sampler = _maybe_sampler(dataset, args)
    return DataLoader(
        dataset, batch_size=bs, shuffle=(sampler is None), sampler=sampler,
        num_workers=0, drop_last=True,
    )


def _nhwc(model, device):
    if device.type == "cuda":
        torch.backends.cudnn.benchmark = True
        return model.to(memory_format=torch.channels_last)
    return model


# ---------------------------------------------------------------------------
# image classification
# ---------------------------------------------------------------------------

def cifar10_main(argv=None, mode=None, client=None, max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("--data_dir", type=str, default=None)
    p.add_argument("--batch_size", type=int, default=128)
    p.add_argument("--lr", type=float, default=0.1)
    common.add_scheduler_args(p, "--num_steps")
    args = p.parse_args(argv)

    criterion = nn.CrossEntropyLoss()

    def step(model, batch, device, state):
        x, y = batch
        x = x.to(device, non_blocking=True)
        if device.type == "cuda":
            x = x.to(memory_format=torch.channels_last)
        y = y.to(device, non_blocking=True)
        return criterion(model(x), y)

    import os as _os

    dataset_len = int(_os.environ.get("SWQ_DATASET_LEN", 50000))
    spec = WorkloadSpec(
        family="ResNet-18",
        build_model=lambda a, d: _nhwc(resnet18_cifar().to(d), d),
        build_loader=lambda a: _loader(
            synthetic.SyntheticImages(dataset_len, 32, 10), a, a.batch_size
        ),
        build_optimizer=lambda a, params: FusedSGD(
            params, lr=a.lr * a.batch_size / 128, momentum=0.9,
            weight_decay=5e-4,
        ),
        step=step,
        make_static_batch=lambda a, d: (
            torch.zeros(a.batch_size, 3, 32, 32, device=d).to(
                memory_format=torch.channels_last
            ),
            torch.zeros(a.batch_size, dtype=torch.long, device=d),
        ),
        copy_batch=lambda s, b: (
            s[0].copy_(b[0], non_blocking=True),
            s[1].copy_(b[1], non_blocking=True),
        ),
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)


def imagenet_main(argv=None, mode=None, client=None, max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("data", nargs="?", default=None)
    p.add_argument("-j", "--workers", type=int, default=4)
    p.add_argument("-a", "--arch", default="resnet50")
    p.add_argument("-b", "--batch_size", type=int, default=64)
    p.add_argument("--lr", type=float, default=0.1)
    common.add_scheduler_args(p, "--num_minibatches")
    args = p.parse_args(argv)

    criterion = nn.CrossEntropyLoss()

    def step(model, batch, device, state):
        x, y = batch
        x = x.to(device, non_blocking=True)
        if device.type == "cuda":
            x = x.to(memory_format=torch.channels_last)
        y = y.to(device, non_blocking=True)
        return criterion(model(x), y)

    spec = WorkloadSpec(
        family="ResNet-50",
        build_model=lambda a, d: _nhwc(resnet50_imagenet().to(d), d),
        build_loader=lambda a: _loader(
            synthetic.SyntheticImages(100000, 224, 1000), a, a.batch_size
        ),
        build_optimizer=lambda a, params: FusedSGD(
            params, lr=a.lr, momentum=0.9, weight_decay=1e-4
        ),
        step=step,
        make_static_batch=lambda a, d: (
            torch.zeros(a.batch_size, 3, 224, 224, device=d).to(
                memory_format=torch.channels_last
            ),
            torch.zeros(a.batch_size, dtype=torch.long, device=d),
        ),
        copy_batch=lambda s, b: (
            s[0].copy_(b[0], non_blocking=True),
            s[1].copy_(b[1], non_blocking=True),
        ),
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)


# ---------------------------------------------------------------------------
# translation (Transformer)
# ---------------------------------------------------------------------------

def translation_main(argv=None, mode=None, client=None,
                     max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("-data", type=str, default=None)
    p.add_argument("-batch_size", type=int, default=64)
    p.add_argument("-proj_share_weight", action="store_true")
    p.add_argument("-lr", type=float, default=1e-4)
    common.add_scheduler_args(p, "-step")
    args = p.parse_args(argv)

    criterion = nn.CrossEntropyLoss(ignore_index=0)

    def step(model, batch, device, state):
        src, tgt = batch
        src = src.to(device, non_blocking=True)
        tgt = tgt.to(device, non_blocking=True)
        out = model(src, tgt[:, :-1])
        return criterion(
            out.reshape(-1, out.size(-1)), tgt[:, 1:].reshape(-1)
        )

    spec = WorkloadSpec(
        family="Transformer",
        build_model=lambda a, d: TranslationTransformer(
            share_proj_weight=a.proj_share_weight
        ).to(d),
        build_loader=lambda a: _loader(
            synthetic.SyntheticTranslation(10000), a, a.batch_size
        ),
        build_optimizer=lambda a, params: FusedAdam(
            params, lr=a.lr, betas=(0.9, 0.98), eps=1e-9
        ),
        step=step,
        supports_accordion=False,   # scheduler rule: no Accordion on
        supports_gns=True,          # transformers (scheduler.py:1670-1672)
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)


# ---------------------------------------------------------------------------
# language modeling (LSTM)
# ---------------------------------------------------------------------------

class _CorpusLoader:
    """bptt-window iterator over the batchified token stream."""

    def __init__(self, corpus, batch_size):
        self.data = corpus.batchify(batch_size)
        self.bptt = corpus.bptt
        self.batch_size = batch_size

    def __len__(self):
        return max(1, (self.data.size(0) - 1) // self.bptt)

    def __iter__(self):
        for i in range(0, self.data.size(0) - 1, self.bptt):
            seq_len = min(self.bptt, self.data.size(0) - 1 - i)
            yield (
                self.data[i : i + seq_len],
                self.data[i + 1 : i + 1 + seq_len].reshape(-1),
            )


def lm_main(argv=None, mode=None, client=None, max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("--cuda", action="store_true")
    p.add_argument("--data", type=str, default=None)
    p.add_argument("--batch_size", type=int, default=20)
    p.add_argument("--lr", type=float, default=20.0)
    p.add_argument("--clip", type=float, default=0.25)
    common.add_scheduler_args(p, "--steps")
    args = p.parse_args(argv)

    criterion = nn.CrossEntropyLoss()

    def step(model, batch, device, state):
        x, y = batch
        x = x.to(device, non_blocking=True)
        y = y.to(device, non_blocking=True)
        module = model.module if hasattr(model, "module") else model
        if "hidden" not in state or state["hidden"][0].size(1) != x.size(1):
            state["hidden"] = module.init_hidden(x.size(1), device)
        state["hidden"] = module.repackage_hidden(state["hidden"])
        out, state["hidden"] = model(x, state["hidden"])
        loss = criterion(out.view(-1, out.size(-1)), y)
        return loss

    spec = WorkloadSpec(
        family="LM",
        build_model=lambda a, d: LSTMLanguageModel().to(d),
        build_loader=lambda a: _CorpusLoader(
            synthetic.SyntheticCorpus(), a.batch_size
        ),
        build_optimizer=lambda a, params: FusedSGD(params, lr=a.lr / 20.0),
        step=step,
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)


# ---------------------------------------------------------------------------
# recommendation (autoencoder)
# ---------------------------------------------------------------------------

def recommendation_main(argv=None, mode=None, client=None,
                        max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("--data_dir", type=str, default=None)
    p.add_argument("--batch_size", type=int, default=2048)
    p.add_argument("--lr", type=float, default=1e-3)
    common.add_scheduler_args(p, "-n")
    args = p.parse_args(argv)

    def step(model, batch, device, state):
        x = batch.to(device, non_blocking=True)
        recon = model(x)
        return RecommendationAutoencoder.loss(recon, x)

    spec = WorkloadSpec(
        family="Recommendation",
        build_model=lambda a, d: RecommendationAutoencoder().to(d),
        build_loader=lambda a: _loader(
            synthetic.SyntheticInteractions(), a, a.batch_size
        ),
        build_optimizer=lambda a, params: FusedAdam(params, lr=a.lr),
        step=step,
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)


# ---------------------------------------------------------------------------
# cyclegan
# ---------------------------------------------------------------------------

def cyclegan_main(argv=None, mode=None, client=None, max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("--dataset_path", type=str, default=None)
    p.add_argument("--decay_epoch", type=int, default=0)
    p.add_argument("--lr", type=float, default=2e-4)
    common.add_scheduler_args(p, "--n_steps")
    args = p.parse_args(argv)
    args.batch_size = 1

    gan_loss = nn.MSELoss()
    cycle_loss = nn.L1Loss()

    class CycleGANBundle(nn.Module):
        def __init__(self):
            super().__init__()
            self.g_ab = GeneratorResNet(num_residual_blocks=6)
            self.g_ba = GeneratorResNet(num_residual_blocks=6)
            self.d_a = Discriminator()
            self.d_b = Discriminator()

    def step(model, batch, device, state):
        m = model.module if hasattr(model, "module") else model
        real_a = batch["A"].to(device, non_blocking=True)
        real_b = batch["B"].to(device, non_blocking=True)
        fake_b = m.g_ab(real_a)
        fake_a = m.g_ba(real_b)
        pred_fake_b = m.d_b(fake_b)
        pred_fake_a = m.d_a(fake_a)
        valid = torch.ones_like(pred_fake_b)
        loss_gan = gan_loss(pred_fake_b, valid) + gan_loss(
            pred_fake_a, torch.ones_like(pred_fake_a)
        )
        loss_cycle = cycle_loss(m.g_ba(fake_b), real_a) + cycle_loss(
            m.g_ab(fake_a), real_b
        )
        return loss_gan + 10.0 * loss_cycle

    spec = WorkloadSpec(
        family="CycleGAN",
        build_model=lambda a, d: CycleGANBundle().to(d),
        build_loader=lambda a: _loader(
            synthetic.SyntheticUnpairedImages(), a, 1
        ),
        build_optimizer=lambda a, params: FusedAdam(
            params, lr=a.lr, betas=(0.5, 0.999)
        ),
        step=step,
        supports_accordion=False,
        supports_gns=False,
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)


# ---------------------------------------------------------------------------
# rl (A3C) — multiprocess Hogwild in the reference; here a single-process
# actor-critic rollout trainer under the lease iterator (rank 0 only, like
# the reference's GavelIterator-on-rank-0 integration, rl/main.py:186-187)
# ---------------------------------------------------------------------------

class _RolloutLoader:
    """Synthetic Pong-like frame rollouts."""

    def __init__(self, n=1000, frame=80):
        self.n = n
        self.frame = frame

    def __len__(self):
        return self.n

    def __iter__(self):
        g = torch.Generator().manual_seed(0)
        for _ in range(self.n):
            yield torch.randn(1, 1, self.frame, self.frame, generator=g)


def rl_main(argv=None, mode=None, client=None, max_steps_override=None):
    p = argparse.ArgumentParser()
    p.add_argument("--env", type=str, default="PongDeterministic-v4")
    p.add_argument("--workers", type=int, default=4,
                   help="Hogwild actor processes; 0 = single-process "
                        "rollout trainer")
    p.add_argument("--rollout", type=int, default=20,
                   help="env frames per actor update (A3C num-steps)")
    p.add_argument("--amsgrad", type=str, default="True")
    p.add_argument("--lr", type=float, default=1e-4)
    common.add_scheduler_args(p, "--max-steps")
    args = p.parse_args(argv)
    args.batch_size = 4

    if args.workers > 0:
        # the reference's default path: lock-free shared-memory actors
        # (rl/main.py:224 torch.multiprocessing + SharedAdam)
        from .hogwild import hogwild_train

        return hogwild_train(
            args, client=client, max_steps_override=max_steps_override
        )

    def step(model, batch, device, state):
        m = model.module if hasattr(model, "module") else model
        x = batch.to(device, non_blocking=True)
        if "hx" not in state:
            state["hx"] = torch.zeros(1, 512, device=device)
            state["cx"] = torch.zeros(1, 512, device=device)
        value, logits, hx, cx = m(x, state["hx"].detach(), state["cx"].detach())
        state["hx"], state["cx"] = hx, cx
        probs = torch.log_softmax(logits, dim=-1)
        # synthetic advantage target
        advantage = torch.randn_like(value)
        policy_loss = -(probs.max(dim=-1).values * advantage.detach()).mean()
        value_loss = 0.5 * (value - advantage).pow(2).mean()
        return policy_loss + value_loss

    spec = WorkloadSpec(
        family="A3C",
        build_model=lambda a, d: ActorCritic().to(d),
        build_loader=lambda a: _RolloutLoader(),
        build_optimizer=lambda a, params: FusedAdam(params, lr=a.lr),
        step=step,
        supports_accordion=False,
        supports_gns=False,
    )
    return run(spec, args, mode=mode, client=client,
               max_steps_override=max_steps_override)
