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

    def test_rl_single_process(self):
        steps = families.rl_main(["--max-steps", "3", "--workers", "0"])
        assert steps == 3

    def test_rl_hogwild(self):
        # the reference's default rl path: torch.multiprocessing actors
        # around a share_memory() model + SharedAdam (rl/main.py:224)
        steps = families.rl_main(
            ["--max-steps", "3", "--workers", "2", "--rollout", "5"]
        )
        assert steps == 3

    def test_rl_hogwild_checkpoint_resume(self, tmp_path):
        args = ["--workers", "2", "--rollout", "5",
                "--checkpoint_dir", str(tmp_path),
                "--enable_gavel_iterator"]
        from shockwave_amd.runtime.lease_iterator import NullLeaseClient

        s1 = families.rl_main(args + ["--max-steps", "2"],
                              client=NullLeaseClient())
        assert s1 == 2
        # second lease resumes from the checkpointed cumulative count
        s2 = families.rl_main(args + ["--max-steps", "5"],
                              client=NullLeaseClient())
        assert s2 == 5  # 2 restored + 3 more


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


class TestRecommendationUnit:
    """Model-level unit tests for the recommendation family — the one
    workload where the reference carries a real pytest suite
    (workloads/pytorch/recommendation/tests/)."""

    def _model(self, n=50):
        from shockwave_amd.models.recommendation import (
            RecommendationAutoencoder,
        )

        return RecommendationAutoencoder(num_items=n, hidden=(16, 8))

    def test_forward_shape_roundtrip(self):
        import torch

        m = self._model(50).eval()
        x = torch.rand(4, 50)
        out = m(x)
        assert out.shape == (4, 50)

    def test_denoising_only_in_training(self):
        import torch

        torch.manual_seed(0)
        m = self._model(50)
        x = torch.rand(8, 50)
        m.eval()
        a, b = m(x), m(x)
        assert torch.equal(a, b)  # no dropout noise in eval

    def test_overfits_tiny_batch(self):
        import torch

        torch.manual_seed(0)
        m = self._model(30)
        m.noise_prob = 0.0
        m.train()
        x = (torch.rand(4, 30) > 0.8).float()
        opt = torch.optim.Adam(m.parameters(), lr=1e-2)
        first = None
        for _ in range(120):
            opt.zero_grad()
            loss = m.loss(m(x), x)
            loss.backward()
            opt.step()
            first = first if first is not None else loss.item()
        assert loss.item() < first * 0.5

    def test_recall_at_k_perfect_and_empty(self):
        import torch

        from shockwave_amd.models.recommendation import recall_at_k

        targets = torch.zeros(2, 10)
        targets[0, [1, 3]] = 1.0
        scores = torch.full((2, 10), -1.0)
        scores[0, 1], scores[0, 3] = 5.0, 4.0
        r = recall_at_k(scores, targets, k=2)
        assert r[0].item() == pytest.approx(1.0)
        assert r[1].item() == pytest.approx(0.0)  # no relevant items

    def test_recall_counts_partial_hits(self):
        import torch

        from shockwave_amd.models.recommendation import recall_at_k

        targets = torch.zeros(1, 10)
        targets[0, [0, 1, 2, 3]] = 1.0
        scores = torch.arange(10, 0, -1).float().unsqueeze(0)
        # top-2 = items 0,1 -> 2 hits / min(4 relevant, 2) = 1.0
        assert recall_at_k(scores, targets, k=2)[0].item() == pytest.approx(1.0)
        # top-8 = items 0..7 -> 4 hits / min(4, 8) = 1.0
        assert recall_at_k(scores, targets, k=8)[0].item() == pytest.approx(1.0)

    def test_ndcg_ordering_sensitivity(self):
        import torch

        from shockwave_amd.models.recommendation import ndcg_at_k

        targets = torch.zeros(1, 6)
        targets[0, 0] = 1.0
        best = torch.tensor([[6.0, 5, 4, 3, 2, 1]])
        worst_in_k = torch.tensor([[1.0, 6, 5, 4, 3, 2]])
        n_best = ndcg_at_k(best, targets, k=3)[0].item()
        n_late = ndcg_at_k(worst_in_k, targets, k=6)[0].item()
        assert n_best == pytest.approx(1.0)
        assert 0 < n_late < n_best


class TestSigtermGracefulPreemption:
    def test_sigterm_checkpoints_and_flushes_progress(self, tmp_path):
        """The dispatcher's watchdog kill sends SIGTERM 10 s before
        SIGKILL; the training loop must catch it, checkpoint, flush
        PROGRESS lines, and exit 0 — so the round's steps are scraped
        instead of lost (the reference SIGKILLs and drops the round)."""
        import signal
        import time

        env = dict(os.environ)
        env.update({
            "SWQ_DATASET_LEN": "50000",
            "SWQ_GRAPHS": "0",
            "GAVEL_JOB_ID": "-1",
            "GAVEL_WORKER_ID": "0",
            "GAVEL_ROUND_ID": "0",
        })
        ckpt = str(tmp_path / "job")
        cmd = [
            sys.executable, "main.py", "--batch_size", "16",
            "--num_steps", "100000", "--checkpoint_dir", ckpt,
            "--enable_gavel_iterator",
        ]
        cwd = os.path.join(REPO, "workloads", "pytorch",
                           "image_classification", "cifar10")
        proc = subprocess.Popen(cmd, cwd=cwd, env=env,
                                stdout=subprocess.PIPE,
                                stderr=subprocess.STDOUT)
        # wait until it has made some steps (first log flush proves the
        # iterator is up), then preempt
        time.sleep(25)
        proc.send_signal(signal.SIGTERM)
        out, _ = proc.communicate(timeout=60)
        assert proc.returncode == 0, out.decode()[-2000:]
        # checkpoint written
        import glob

        assert glob.glob(os.path.join(ckpt, "*.chkpt")) or glob.glob(
            os.path.join(ckpt, "*.pt")
        ) or glob.glob(os.path.join(ckpt, "model*")), os.listdir(ckpt)
        # progress lines flushed with nonzero steps
        log = os.path.join(ckpt, ".gavel", "round=0", "worker=0.log")
        steps = 0
        for line in open(log):
            if "[PROGRESS] [STEPS]" in line:
                steps = int(float(line.rsplit("]", 1)[1].strip()))
        assert steps > 0, open(log).read()[-1500:]


class TestModelShapes:
    """Direct model-level shape/gradient checks for every family
    (complements the training-loop tests above)."""

    def test_resnet50_imagenet_shapes(self):
        import torch

        from shockwave_amd.models import resnet50_imagenet

        m = resnet50_imagenet()
        out = m(torch.randn(2, 3, 224, 224))
        assert out.shape == (2, 1000)

    def test_transformer_shapes_and_grad(self):
        import torch

        from shockwave_amd.models.transformer import TranslationTransformer

        m = TranslationTransformer(src_vocab=100, tgt_vocab=90,
                                   d_model=32, nhead=4, num_layers=1)
        src = torch.randint(0, 100, (3, 7))
        tgt = torch.randint(0, 90, (3, 5))
        out = m(src, tgt)
        assert out.shape[:2] == (3, 5) and out.shape[2] == 90
        out.sum().backward()
        assert any(p.grad is not None for p in m.parameters())

    def test_lstm_lm_hidden_carry(self):
        import torch

        from shockwave_amd.models.lstm_lm import LSTMLanguageModel

        m = LSTMLanguageModel(vocab=120, emsize=16, nhid=16, nlayers=1)
        h = m.init_hidden(2, torch.device("cpu"))
        x = torch.randint(0, 120, (5, 2))
        out, h2 = m(x, h)
        assert out.shape[-1] == 120
        assert h2[0].shape == h[0].shape

    def test_cyclegan_generator_roundtrip(self):
        import torch

        from shockwave_amd.models.cyclegan import (
            Discriminator, GeneratorResNet,
        )

        g = GeneratorResNet(channels=3, num_residual_blocks=1)
        x = torch.randn(1, 3, 64, 64)
        assert g(x).shape == (1, 3, 64, 64)
        d = Discriminator(channels=3)
        assert d(x).shape[0] == 1

    def test_a3c_policy_value_heads(self):
        import torch

        from shockwave_amd.models.a3c import ActorCritic

        m = ActorCritic(num_inputs=1, num_actions=6)
        x = torch.randn(1, 1, 80, 80)
        hx = torch.zeros(1, m.lstm.hidden_size)
        cx = torch.zeros(1, m.lstm.hidden_size)
        value, logits, hx2, cx2 = m(x, hx, cx)
        assert logits.shape == (1, 6)
        assert value.shape == (1, 1)
        assert hx2.shape == hx.shape
