"""Session-cache correctness: job state must be fully isolated from the
cached executor (model/optimizer/adaptation state restored in place per
lease), or a warm runner would leak one job's training into another."""

import os

import pytest
import torch

from shockwave_amd.workloads import families, session


@pytest.fixture(autouse=True)
def _clear_sessions():
    session.clear()
    yield
    session.clear()


def run_cifar(tmp_path, job, steps, bs=16):
    ckpt = str(tmp_path / f"job_id={job}")
    argv = [
        "--batch_size", str(bs), "--num_steps", str(steps),
        "--checkpoint_dir", ckpt, "--enable_gavel_iterator",
    ]
    from shockwave_amd.runtime.lease_iterator import NullLeaseClient

    return families.cifar10_main(argv, mode="static",
                                 client=NullLeaseClient())


def model_state(tmp_path, job):
    ckpt = torch.load(
        str(tmp_path / f"job_id={job}" / "model.chkpt"),
        map_location="cpu", weights_only=False,
    )
    return ckpt


class TestSessionReuse:
    def test_same_config_shares_executor(self, tmp_path):
        run_cifar(tmp_path, job=0, steps=2)
        assert len(session._CACHE) == 1
        sess = next(iter(session._CACHE.values()))
        model_a = sess.model
        run_cifar(tmp_path, job=1, steps=2)
        assert len(session._CACHE) == 1
        assert next(iter(session._CACHE.values())).model is model_a
        assert sess.uses >= 1

    def test_different_bs_different_session(self, tmp_path):
        run_cifar(tmp_path, job=0, steps=2, bs=16)
        run_cifar(tmp_path, job=1, steps=2, bs=32)
        assert len(session._CACHE) == 2

    def test_fresh_job_does_not_inherit_weights(self, tmp_path):
        """Job B (no checkpoint) reusing job A's session must behave
        exactly like a job in a cold process: same weights AND optimizer
        state after one step (job state fully isolated from A's)."""
        torch.manual_seed(123)
        run_cifar(tmp_path, job=0, steps=4)
        sess = next(iter(session._CACHE.values()))
        init = sess.init_model_state
        trained = sess.model.state_dict()
        moved = sum(
            (trained[k].float() - init[k].to(trained[k].device).float())
            .abs().sum().item() for k in init
        )
        assert moved > 0  # A actually trained

        torch.manual_seed(777)
        run_cifar(tmp_path, job=1, steps=1)  # warm reuse, fresh job
        warm = model_state(tmp_path, 1)

        session.clear()
        torch.manual_seed(123)  # reproduce the session-creation RNG state
        run_cifar(tmp_path, job=5, steps=4)  # rebuild equivalent session
        torch.manual_seed(777)
        run_cifar(tmp_path, job=2, steps=1)  # cold-equivalent fresh job
        cold = model_state(tmp_path, 2)

        for k in warm["model"]:
            torch.testing.assert_close(
                warm["model"][k], cold["model"][k],
                msg=lambda m: f"model {k}: {m}",
            )
        w_opt = warm["optimizer"]["state"]
        c_opt = cold["optimizer"]["state"]
        assert w_opt.keys() == c_opt.keys()
        for k in w_opt:
            for name, v in w_opt[k].items():
                if torch.is_tensor(v):
                    torch.testing.assert_close(
                        v, c_opt[k][name],
                        msg=lambda m: f"opt {k}.{name}: {m}",
                    )

    def test_resume_restores_job_state(self, tmp_path):
        """A -> B -> A again: A's second lease resumes from A's
        checkpoint exactly, despite B trampling the shared model."""
        run_cifar(tmp_path, job=0, steps=3)
        a_ckpt = model_state(tmp_path, 0)
        run_cifar(tmp_path, job=1, steps=3)
        # resume A with 0 additional steps: saved state == prior ckpt
        run_cifar(tmp_path, job=0, steps=3)  # target already reached
        a_after = model_state(tmp_path, 0)
        for k in a_ckpt["model"]:
            torch.testing.assert_close(
                a_after["model"][k], a_ckpt["model"][k],
                msg=lambda m: f"{k}: {m}",
            )
        assert a_after["cumulative_steps"] == a_ckpt["cumulative_steps"]

    def test_disabled_by_env(self, tmp_path, monkeypatch):
        monkeypatch.setenv("SWQ_SESSION_CACHE", "0")
        run_cifar(tmp_path, job=0, steps=1)
        assert len(session._CACHE) == 0

    def test_eviction_bounded(self, tmp_path, monkeypatch):
        monkeypatch.setenv("SWQ_SESSION_CACHE_SIZE", "2")
        run_cifar(tmp_path, job=0, steps=1, bs=16)
        run_cifar(tmp_path, job=1, steps=1, bs=32)
        run_cifar(tmp_path, job=2, steps=1, bs=64)
        assert len(session._CACHE) == 2
