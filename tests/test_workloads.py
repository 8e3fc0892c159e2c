"""Workload family smoke tests (CPU, few steps each)."""

import subprocess
import sys
import os

import pytest

from shockwave_amd.workloads import families

REPO = os.path.join(os.path.dirname(__file__), "..")


class TestFamilies:
    def test_cifar10(self):
        steps = families.cifar10_main(
            ["--batch_size", "16", "--num_steps", "3"]
        )
        assert steps == 3

    @pytest.mark.slow
    def test_imagenet(self):
        steps = families.imagenet_main(["-b", "4", "--num_minibatches", "2"])
        assert steps == 2

    def test_translation(self):
        steps = families.translation_main(
            ["-batch_size", "8", "-proj_share_weight", "-step", "2"]
        )
        assert steps == 2

    def test_lm(self):
        steps = families.lm_main(["--batch_size", "10", "--steps", "3"])
        assert steps == 3

    def test_recommendation(self):
        steps = families.recommendation_main(
            ["--batch_size", "64", "-n", "3"]
        )
        assert steps == 3

    @pytest.mark.slow
    def test_cyclegan(self):
        steps = families.cyclegan_main(["--n_steps", "1"])
        assert steps == 1

    def test_rl(self):
        steps = families.rl_main(["--max-steps", "3"])
        assert steps == 3


class TestShims:
    @pytest.mark.parametrize("tree", ["pytorch", "accordion", "gns"])
    def test_cifar10_shim_runs(self, tree, tmp_path):
        cmd = [
            sys.executable, "main.py", "--batch_size", "16",
            "--num_steps", "2",
        ]
        cwd = os.path.join(REPO, "workloads", tree,
                           "image_classification", "cifar10")
        r = subprocess.run(cmd, cwd=cwd, capture_output=True, timeout=300)
        assert r.returncode == 0, r.stdout.decode() + r.stderr.decode()


class TestLeaseIntegration:
    def test_cifar10_lease_preemption_and_resume(self, tmp_path):
        """Job runs under a finite lease, checkpoints, and resumes with
        preserved step count — the cooperative-preemption contract."""
        from tests.test_rpc_runtime import FakeLeaseClient

        ckpt_dir = str(tmp_path)
        client = FakeLeaseClient([(4, 1e9)])
        steps1 = families.cifar10_main(
            ["--batch_size", "16", "--num_steps", "100",
             "--checkpoint_dir", ckpt_dir, "--enable_gavel_iterator"],
            client=client,
        )
        assert steps1 == 4  # preempted by lease
        assert os.path.exists(os.path.join(ckpt_dir, "model.chkpt"))

        client2 = FakeLeaseClient([(4, 1e9)])
        steps2 = families.cifar10_main(
            ["--batch_size", "16", "--num_steps", "100",
             "--checkpoint_dir", ckpt_dir, "--enable_gavel_iterator"],
            client=client2,
        )
        # resumed from checkpoint: cumulative steps continue from 4
        assert steps2 == 8

    def test_accordion_requests_rescale(self, tmp_path):
        """Accordion mode leaves the critical regime -> requests big_bs."""
        from tests.test_rpc_runtime import FakeLeaseClient

        client = FakeLeaseClient([(int(1e9), 1e9)])
        import shockwave_amd.workloads.loop as loop_mod
        from shockwave_amd.data.synthetic import SyntheticImages

        # tiny dataset so epochs are fast; run enough epochs to leave the
        # critical regime (>30% of training AND epoch >= 10)
        orig = SyntheticImages.__len__
        try:
            SyntheticImages.__len__ = lambda self: 16 * 3  # 3 steps/epoch
            steps = families.cifar10_main(
                ["--batch_size", "16", "--num_steps", "200",
                 "--checkpoint_dir", str(tmp_path),
                 "--enable_gavel_iterator", "--mode", "accordion"],
                client=client,
            )
        finally:
            SyntheticImages.__len__ = orig
        assert client.rr_calls, "accordion should have requested a rescale"
        assert client.rr_calls[0] == (True, False)


class TestGNSLoopIntegration:
    def test_gns_double_requested_when_estimator_fires(self, tmp_path,
                                                       monkeypatch):
        """When GNS says double, the loop reports big_bs and checkpoints."""
        from tests.test_rpc_runtime import FakeLeaseClient
        from shockwave_amd.adapt.gns import GNSEstimator

        monkeypatch.setattr(GNSEstimator, "should_double",
                            lambda self, epoch, lookback=10: True)
        import os
        os.environ["SWQ_DATASET_LEN"] = "64"  # 4 steps/epoch at bs16
        try:
            client = FakeLeaseClient([(int(1e9), 1e9)])
            steps = families.cifar10_main(
                ["--batch_size", "16", "--num_steps", "50",
                 "--checkpoint_dir", str(tmp_path),
                 "--enable_gavel_iterator", "--mode", "gns"],
                client=client,
            )
            assert client.rr_calls == [(True, False)]
            # synthetic-data mode ends the epoch by raising on the 4th
            # __next__ (after counting it), so the loop body ran 3 times —
            # reference GavelIterator semantics (gavel_iterator.py:160-173)
            assert steps == 3
        finally:
            del os.environ["SWQ_DATASET_LEN"]


class TestAccordionScaleDown:
    def test_reentering_critical_regime_requests_small_bs(self, tmp_path):
        """A job already scaled up (bs 256, original 32) that re-enters the
        critical regime (epochs 150-159 for ResNet-18) reports small_bs."""
        from tests.test_rpc_runtime import FakeLeaseClient
        from shockwave_amd.parallel.ckpt_stream import CheckpointStore
        from shockwave_amd.models import resnet18_cifar
        from shockwave_amd.ops.optim import FusedSGD
        import os
        import torch

        # fabricate the checkpoint a previous (scaled-up) lease would leave
        model = resnet18_cifar()
        opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9)
        store = CheckpointStore(str(tmp_path))
        store.save({
            "model": model.state_dict(),
            "optimizer": opt.state_dict(),
            "epoch": 150,           # inside the 150-159 critical window
            "cumulative_steps": 0,
            "original_bs": 32,
        })

        os.environ["SWQ_DATASET_LEN"] = "512"  # 2 steps/epoch at bs 256
        try:
            client = FakeLeaseClient([(int(1e9), 1e9)])
            families.cifar10_main(
                ["--batch_size", "256", "--num_steps", "50",
                 "--checkpoint_dir", str(tmp_path),
                 "--enable_gavel_iterator", "--mode", "accordion"],
                client=client,
            )
            assert client.rr_calls == [(False, True)]
        finally:
            del os.environ["SWQ_DATASET_LEN"]
