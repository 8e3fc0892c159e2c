"""Hogwild (lock-free shared-memory) A3C trainer.

MI355X-native rebuild of the reference's multiprocess rl workload
(reference workloads/pytorch/rl/main.py:224: ``torch.multiprocessing``
actor processes around a ``share_memory()`` model + SharedAdam, as in
pytorch-a3c).  Design here:

* the SHARED model and Adam moments live in CPU shared memory (the
  canonical Hogwild layout — parameters must be lock-free writable by
  every actor, and HIP device memory is per-process);
* each actor process runs its rollouts' forward/backward on the MI355X
  (all actors time-share the one visible GPU — the same co-residency
  the dispatcher uses for packed jobs), then applies its gradients to
  the shared parameters without locks (last-writer-wins, Hogwild);
* the parent process owns the lease: one LeaseIterator step per global
  optimizer update (a shared counter), so scheduling, checkpointing
  and preemption work exactly as for every other family.

The single-process rollout trainer (families.rl_main --workers 0)
remains for environments where process spawning is unavailable.
"""

from __future__ import annotations

import time

import torch
import torch.multiprocessing as mp


class SharedAdam(torch.optim.Adam):
    """Adam whose state tensors live in shared memory before any step,
    so every Hogwild actor updates the same moments (pytorch-a3c
    SharedAdam pattern, rebuilt on the stock torch 2.x optimizer)."""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=0.0, amsgrad=True):
        super().__init__(params, lr=lr, betas=betas, eps=eps,
                         weight_decay=weight_decay, amsgrad=amsgrad)
        for group in self.param_groups:
            for p in group["params"]:
                state = self.state[p]
                # mirror stock Adam's lazy init exactly, then share
                state["step"] = torch.tensor(0.0)
                state["exp_avg"] = torch.zeros_like(
                    p, memory_format=torch.preserve_format
                )
                state["exp_avg_sq"] = torch.zeros_like(
                    p, memory_format=torch.preserve_format
                )
                if amsgrad:
                    state["max_exp_avg_sq"] = torch.zeros_like(
                        p, memory_format=torch.preserve_format
                    )
                for t in state.values():
                    t.share_memory_()


def synthetic_rollout(local, device, state, rollout_len, gen):
    """One A3C rollout on synthetic Pong-like frames: T LSTM-carried
    steps, policy + value + entropy losses against synthetic returns
    (no network access for real Atari; the compute shape matches)."""
    hx, cx = state
    values, log_probs, entropies = [], [], []
    for _ in range(rollout_len):
        x = torch.randn(1, 1, 80, 80, generator=gen).to(device)
        value, logits, hx, cx = local(x, hx, cx)
        prob = torch.softmax(logits, dim=-1)
        log_prob = torch.log_softmax(logits, dim=-1)
        entropies.append(-(log_prob * prob).sum(1))
        action = prob.multinomial(num_samples=1).detach()
        log_probs.append(log_prob.gather(1, action))
        values.append(value)
    returns = torch.randn(len(values), generator=gen).to(device)
    policy_loss = value_loss = 0
    for i in range(len(values)):
        advantage = returns[i] - values[i]
        value_loss = value_loss + 0.5 * advantage.pow(2)
        policy_loss = (
            policy_loss
            - log_probs[i] * advantage.detach()
            - 0.01 * entropies[i]
        )
    return (policy_loss + 0.5 * value_loss).sum(), (hx.detach(), cx.detach())


def _actor(rank, shared, opt, counter, stop, rollout_len, use_cuda):
    from ..models import ActorCritic

    torch.manual_seed(1000 + rank)
    device = (
        torch.device("cuda", 0)
        if use_cuda and torch.cuda.is_available()
        else torch.device("cpu")
    )
    local = ActorCritic().to(device)
    gen = torch.Generator().manual_seed(rank)
    state = (
        torch.zeros(1, 512, device=device),
        torch.zeros(1, 512, device=device),
    )
    while not stop.is_set():
        local.load_state_dict(shared.state_dict())
        loss, state = synthetic_rollout(
            local, device, state, rollout_len, gen
        )
        local.zero_grad()
        loss.backward()
        torch.nn.utils.clip_grad_norm_(local.parameters(), 40.0)
        for sp, lp in zip(shared.parameters(), local.parameters()):
            if lp.grad is None:
                continue
            g = lp.grad.detach().to("cpu")
            if sp.grad is None:
                sp.grad = g  # this process's private grad slot
            else:
                sp.grad.copy_(g)
        opt.step()  # lock-free on the SHARED params/moments (Hogwild)
        with counter.get_lock():
            counter.value += 1


def hogwild_train(args, client=None, max_steps_override=None):
    """Parent: owns the lease; children: Hogwild actors.  Returns the
    number of global optimizer updates run this lease."""
    from ..models import ActorCritic
    from . import common

    shared = ActorCritic()
    shared.share_memory()
    opt = SharedAdam(
        shared.parameters(), lr=args.lr,
        amsgrad=str(args.amsgrad) == "True",
    )

    target_steps = max_steps_override or args.num_steps or 400

    class _Ticks:
        def __len__(self):
            return 1000

        def __iter__(self):
            return iter(range(1000))

    trainloader, lease_it = common.make_lease_iterator(
        _Ticks(), args, synthetic_data=True, client=client
    )

    cumulative = 0
    ckpt = lease_it.load_checkpoint() if lease_it is not None else None
    if ckpt:
        shared.load_state_dict(ckpt["model"])
        try:
            opt.load_state_dict(ckpt["optimizer"])
            for group in opt.param_groups:
                for p in group["params"]:
                    for t in opt.state[p].values():
                        if torch.is_tensor(t):
                            t.share_memory_()
        except (ValueError, KeyError):
            pass
        cumulative = ckpt.get("cumulative_steps", 0)

    ctx = mp.get_context("spawn")
    counter = ctx.Value("l", 0)
    stop = ctx.Event()
    actors = [
        ctx.Process(
            target=_actor,
            args=(r, shared, opt, counter, stop, args.rollout,
                  torch.cuda.is_available()),
            daemon=True,
        )
        for r in range(args.workers)
    ]
    for a in actors:
        a.start()

    seen = 0
    try:
        for _ in trainloader:
            # one lease step == one global optimizer update
            deadline = time.time() + 120.0
            while counter.value <= seen:
                time.sleep(0.002)
                if all(not a.is_alive() for a in actors):
                    raise RuntimeError("all Hogwild actors died")
                if time.time() > deadline:
                    raise RuntimeError("Hogwild actors stalled")
            seen += 1
            cumulative += 1
            if cumulative >= target_steps:
                break
            if lease_it is not None and lease_it.done:
                break
    finally:
        stop.set()
        for a in actors:
            a.join(timeout=30)
            if a.is_alive():
                a.terminate()

    if lease_it is not None:
        lease_it.save_checkpoint({
            "model": shared.state_dict(),
            "optimizer": opt.state_dict(),
            "cumulative_steps": cumulative,
        })
        if cumulative >= target_steps and not lease_it.done:
            lease_it.complete()
        lease_it.write_progress()
        lease_it.close()
    return cumulative
