"""Checkpoint-store tests: RAM tier, durability, remote fetch."""

import os

import pytest
import torch

from shockwave_amd.parallel.ckpt_stream import CheckpointStore, fetch_remote_checkpoint


class TestCheckpointStore:
    def test_save_load_roundtrip(self, tmp_path):
        store = CheckpointStore(str(tmp_path / "job_id=1"),
                                shm_root=str(tmp_path / "shm"))
        state = {"w": torch.randn(100), "step": 42,
                 "nested": {"b": torch.ones(3)}}
        store.save(state)
        loaded = store.load()
        assert loaded["step"] == 42
        torch.testing.assert_close(loaded["w"], state["w"])
        torch.testing.assert_close(loaded["nested"]["b"], state["nested"]["b"])

    def test_shm_tier_preferred_but_durable_survives(self, tmp_path):
        store = CheckpointStore(str(tmp_path / "job_id=2"),
                                shm_root=str(tmp_path / "shm"))
        store.save({"x": torch.tensor([1.0])})
        # simulate node restart: RAM tier gone
        store.clear()
        loaded = store.load()
        assert loaded is not None
        assert float(loaded["x"]) == 1.0

    def test_bytes_roundtrip(self, tmp_path):
        store = CheckpointStore(str(tmp_path / "job_id=3"),
                                shm_root=str(tmp_path / "shm"))
        store.save({"x": torch.arange(10)})
        data = store.read_bytes()
        store2 = CheckpointStore(str(tmp_path / "job_id=3b"),
                                 shm_root=str(tmp_path / "shm"))
        store2.write_bytes(data)
        assert torch.equal(store2.load()["x"], torch.arange(10))


class TestRemoteFetch:
    def test_fetch_over_worker_rpc(self, tmp_path):
        from shockwave_amd.rpc.services import serve_worker
        from shockwave_amd.rpc.transport import RpcClient
        from tests.test_rpc_runtime import free_port

        ckpt_root = tmp_path / "ckpts"
        store = CheckpointStore(str(ckpt_root / "job_id=7"),
                                shm_root=str(tmp_path / "shm"))
        payload = {"w": torch.randn(50000)}  # ~200 KB
        store.save(payload)

        def fetch(job_id, offset, length):
            s = CheckpointStore(str(ckpt_root / f"job_id={job_id}"),
                                shm_root=str(tmp_path / "shm"))
            data = s.read_bytes()
            if data is None:
                return {"found": False, "data": b"", "total": 0}
            return {"found": True, "data": data[offset:offset + length],
                    "total": len(data)}

        port = free_port()
        server = serve_worker(port, {
            "RunJob": lambda *a: None, "KillJob": lambda *a: None,
            "Reset": lambda: None, "Shutdown": lambda: None,
            "FetchCheckpoint": fetch,
        })
        try:
            client = RpcClient("127.0.0.1", port)
            data = fetch_remote_checkpoint(client, 7)
            assert data == store.read_bytes()
            data_missing = fetch_remote_checkpoint(client, 99)
            assert data_missing is None
        finally:
            server.stop(0)


class TestCheckpointStoreEdgeCases:
    def test_load_missing_returns_none(self, tmp_path):
        store = CheckpointStore(str(tmp_path / "job_id=9"),
                                shm_root=str(tmp_path / "shm"))
        assert store.load() is None
        assert store.read_bytes() is None

    def test_stale_shm_cache_invalidated_by_newer_durable(self, tmp_path):
        """The shm tier is a strict mtime-validated cache: a NEWER durable
        file written by another store instance (e.g. after a migration)
        must win over a stale shm copy."""
        import time

        a = CheckpointStore(str(tmp_path / "job_id=10"),
                            shm_root=str(tmp_path / "shm"))
        a.save({"v": torch.tensor([1.0])})
        time.sleep(0.05)
        # a different process/instance updates the durable checkpoint
        b = CheckpointStore(str(tmp_path / "job_id=10"),
                            shm_root=str(tmp_path / "other_shm"))
        b.save({"v": torch.tensor([2.0])})
        # original instance (stale shm) must pick up the newer durable
        assert float(a.load()["v"]) == 2.0

    def test_corrupt_durable_does_not_crash(self, tmp_path):
        store = CheckpointStore(str(tmp_path / "job_id=11"),
                                shm_root=str(tmp_path / "shm"))
        store.save({"v": torch.tensor([3.0])})
        store.clear()  # drop shm
        # corrupt the durable file
        for name in os.listdir(tmp_path / "job_id=11"):
            p = tmp_path / "job_id=11" / name
            if p.is_file():
                p.write_bytes(b"garbage")
        try:
            out = store.load()
        except Exception as e:
            pytest.fail(f"corrupt checkpoint must not raise: {e!r}")
        assert out is None

    def test_read_bytes_skips_torn_shm_tier(self, tmp_path):
        """a torn /dev/shm copy must not be served over
        FetchCheckpoint when the durable copy is intact."""
        import io

        store = CheckpointStore(str(tmp_path / "job_id=12"),
                                shm_root=str(tmp_path / "shm"))
        store.save({"v": torch.tensor([4.0])})
        # tear the shm copy (newer mtime than durable so it is preferred)
        shm_path = store._paths()[0]
        with open(shm_path, "r+b") as f:
            f.truncate(16)
        os.utime(shm_path)
        data = store.read_bytes()
        assert data is not None
        out = torch.load(io.BytesIO(data), weights_only=False)
        assert float(out["v"]) == 4.0
