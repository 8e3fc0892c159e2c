"""Distributed data-parallel tests over gloo (world_size 2, CPU)."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp


def _ddp_worker(rank, world_size, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        from shockwave_amd.parallel import BucketedDataParallel

        torch.manual_seed(100 + rank)  # different init per rank
        model = torch.nn.Sequential(
            torch.nn.Linear(16, 64), torch.nn.ReLU(), torch.nn.Linear(64, 4)
        )
        ddp = BucketedDataParallel(model, bucket_bytes=1024)

        # after wrap, params must match rank 0's
        psum = sum(p.sum().item() for p in model.parameters())

        torch.manual_seed(rank)  # different data per rank
        x = torch.randn(8, 16)
        y = torch.randint(0, 4, (8,))
        out = ddp(x)
        loss = torch.nn.functional.cross_entropy(out, y)
        loss.backward()
        ddp.finish_gradient_sync()
        gsum = sum(
            p.grad.sum().item() for p in model.parameters() if p.grad is not None
        )

        # reference: manually averaged gradients of both ranks
        model_ref = torch.nn.Sequential(
            torch.nn.Linear(16, 64), torch.nn.ReLU(), torch.nn.Linear(64, 4)
        )
        model_ref.load_state_dict(model.state_dict())
        grads = []
        for r in range(world_size):
            torch.manual_seed(r)
            xr = torch.randn(8, 16)
            yr = torch.randint(0, 4, (8,))
            model_ref.zero_grad()
            torch.nn.functional.cross_entropy(model_ref(xr), yr).backward()
            grads.append([p.grad.clone() for p in model_ref.parameters()])
        avg = [sum(g[i] for g in grads) / world_size for i in range(len(grads[0]))]
        max_err = max(
            (a - p.grad).abs().max().item()
            for a, p in zip(avg, model.parameters())
        )
        q.put((rank, psum, gsum, max_err))
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, "ERROR", traceback.format_exc(), str(e)))


class TestBucketedDDP:
    def test_gradient_averaging_world2(self):
        import socket

        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()

        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        procs = [
            ctx.Process(target=_ddp_worker, args=(r, 2, port, q))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        results = [q.get() for _ in range(2)]
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0
        for r in results:
            assert r[1] != "ERROR", r[2]
        # both ranks saw the same (broadcast) params
        assert results[0][1] == pytest.approx(results[1][1], rel=1e-6)
        # both ranks ended with identical averaged gradients
        assert results[0][2] == pytest.approx(results[1][2], rel=1e-6)
        # and those equal the manual average of per-rank gradients
        for r in results:
            assert r[3] < 1e-6


class TestBucketViews:
    def test_grads_are_views_into_buckets(self):
        import torch.distributed as dist

        if not dist.is_initialized():
            store = dist.TCPStore("127.0.0.1", 0, 1, True)
            dist.init_process_group(
                "gloo", store=store, rank=0, world_size=1
            )
        from shockwave_amd.parallel import BucketedDataParallel

        model = torch.nn.Linear(8, 8)
        ddp = BucketedDataParallel(model, bucket_bytes=16)
        x = torch.randn(4, 8)
        ddp(x).sum().backward()
        ddp.finish_gradient_sync()
        for p in model.parameters():
            assert p.grad is not None
            # grad storage belongs to one of the bucket buffers
            assert any(
                p.grad.data_ptr() >= b.data_ptr()
                and p.grad.data_ptr() < b.data_ptr() + b.numel() * b.element_size()
                for b in ddp.grad_buffers
            )
        ddp.zero_grad()
        assert all(b.abs().sum() == 0 for b in ddp.grad_buffers)


class TestManualSyncMode:
    def test_manual_sync_world1(self):
        import torch.distributed as dist

        if not dist.is_initialized():
            store = dist.TCPStore("127.0.0.1", 0, 1, True)
            dist.init_process_group("gloo", store=store, rank=0, world_size=1)
        from shockwave_amd.parallel import BucketedDataParallel

        torch.manual_seed(3)
        model = torch.nn.Linear(8, 4)
        ddp = BucketedDataParallel(model, bucket_bytes=64)
        ddp.sync_mode = "manual"
        x = torch.randn(4, 8)
        ddp(x).sum().backward()
        g_before = [b.clone() for b in ddp.grad_buffers]
        ddp.finish_gradient_sync()  # world 1: average is a no-op
        for b, g in zip(ddp.grad_buffers, g_before):
            torch.testing.assert_close(b, g)
        # calling again (replay-driven cadence) must not error
        ddp.finish_gradient_sync()


class TestBucketAlignment:
    def test_views_are_16B_aligned_with_odd_numels(self):
        """params with numel%4!=0 (e.g. a bias of 10) must not
        misalign the following grads — the fused kernels cast to float4*."""
        import torch.distributed as dist

        if not dist.is_initialized():
            store = dist.TCPStore("127.0.0.1", 0, 1, True)
            dist.init_process_group("gloo", store=store, rank=0, world_size=1)
        from shockwave_amd.parallel import BucketedDataParallel

        model = torch.nn.Sequential(
            torch.nn.Linear(7, 10),   # weight 70, bias 10 — both odd
            torch.nn.Linear(10, 3),   # weight 30, bias 3
        )
        ddp = BucketedDataParallel(model, bucket_bytes=10**9)  # one bucket
        for p in model.parameters():
            assert p.grad.data_ptr() % 16 == 0, p.shape
        # padding must not break correctness
        x = torch.randn(4, 7)
        ddp(x).sum().backward()
        ddp.finish_gradient_sync()
        ref = torch.nn.Sequential(
            torch.nn.Linear(7, 10), torch.nn.Linear(10, 3)
        )
        ref.load_state_dict(model.state_dict())
        ref(x).sum().backward()
        for p, r in zip(model.parameters(), ref.parameters()):
            torch.testing.assert_close(p.grad, r.grad)

    def test_multi_forward_single_sync(self):
        """GAN-style: two forwards, two backwards, then one sync — the
        second forward must not reset the countdown mid-flight."""
        import torch.distributed as dist

        if not dist.is_initialized():
            store = dist.TCPStore("127.0.0.1", 0, 1, True)
            dist.init_process_group("gloo", store=store, rank=0, world_size=1)
        from shockwave_amd.parallel import BucketedDataParallel

        torch.manual_seed(7)
        model = torch.nn.Linear(6, 6)
        ddp = BucketedDataParallel(model, bucket_bytes=16)
        xa, xb = torch.randn(3, 6), torch.randn(3, 6)
        la = ddp(xa).sum()
        lb = ddp(xb).sum()    # second forward before any backward
        la.backward()
        lb.backward()
        ddp.finish_gradient_sync()
        ref = torch.nn.Linear(6, 6)
        ref.load_state_dict(model.state_dict())
        (ref(xa).sum() + ref(xb).sum()).backward()
        for p, r in zip(model.parameters(), ref.parameters()):
            torch.testing.assert_close(p.grad, r.grad)


def _agree_worker(rank, world_size, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        from shockwave_amd.parallel.graphs import agree_capture

        sentinel = object()
        # round 1: only rank 0 "captured" -> everyone must discard
        mine = sentinel if rank == 0 else None
        r1 = agree_capture(mine)
        # round 2: every rank captured -> everyone keeps its graph
        r2 = agree_capture(sentinel)
        q.put((rank, r1 is None, r2 is sentinel))
        dist.destroy_process_group()
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, "ERROR", traceback.format_exc()))


class TestAgreeCapture:
    """All ranks must converge on one hipGraph mode (bench.py / loop.py
    wire this in at world>1 so captured RCCL schedules never mix with
    eager fallbacks across ranks)."""

    def test_world2_agreement(self):
        import socket

        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()

        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        procs = [
            ctx.Process(target=_agree_worker, args=(r, 2, port, q))
            for r in range(2)
        ]
        for p in procs:
            p.start()
        results = [q.get() for _ in range(2)]
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0
        for r in results:
            assert r[1] != "ERROR", r[2]
            assert r[1] is True   # disagreement -> discarded everywhere
            assert r[2] is True   # unanimity -> kept

    def test_world1_passthrough(self):
        from shockwave_amd.parallel.graphs import agree_capture

        s = object()
        assert agree_capture(s) is s  # no process group: no-op
