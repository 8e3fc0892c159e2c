"""Wire-format protobuf compatibility (rpc/pb.py runtime stubs).

The reference builds its stubs with protoc; this image has only the
google.protobuf runtime, so rpc/pb.py parses the committed .proto files
and materializes real message classes at runtime.  These tests prove
(1) the classes round-trip, (2) their bytes interoperate with classes
built from the REFERENCE's own .proto files (the actual wire-compat
claim), and (3) the full RPC stack works with SWQ_RPC_CODEC=proto.
"""

import os
import sys

import pytest

REPO = os.path.join(os.path.dirname(__file__), "..")
sys.path.insert(0, REPO)

REF_PROTOS = "/root/reference/scheduler/runtime/protobuf"

from shockwave_amd.rpc.pb import Schema, our_schema  # noqa: E402


class TestRuntimeStubs:
    def test_all_messages_materialize(self):
        s = our_schema()
        for name in [
            "Empty", "JobState", "RegisterWorkerRequest",
            "RegisterWorkerResponse", "Heartbeat", "DoneRequest",
            "JobDescription", "RunJobRequest", "KillJobRequest",
            "FetchCheckpointRequest", "FetchCheckpointResponse",
            "InitJobRequest", "UpdateLeaseRequest", "UpdateLeaseResponse",
            "UpdateResourceRequirementRequest",
        ]:
            assert name in s.messages

    def test_roundtrip_nested_repeated(self):
        s = our_schema()
        payload = {
            "job_descriptions": [
                {"job_id": 7, "job_type": "ResNet-18 (batch size 16)",
                 "command": "python3 main.py", "working_directory": "x",
                 "needs_data_dir": True, "num_steps_arg": "--num_steps",
                 "num_steps": 400, "has_duration": False, "duration": 0,
                 "mode": "static", "mps_thread_percentage": 100},
            ],
            "worker_id": 3,
            "round_id": 12,
        }
        m = s.to_message("RunJobRequest", payload)
        m2 = s.messages["RunJobRequest"]()
        m2.ParseFromString(m.SerializeToString())
        d = s.to_dict(m2)
        assert d["worker_id"] == 3 and d["round_id"] == 12
        jd = d["job_descriptions"][0]
        assert jd["job_id"] == 7 and jd["num_steps"] == 400
        assert jd["mode"] == "static"

    def test_bytes_field(self):
        s = our_schema()
        m = s.to_message(
            "FetchCheckpointResponse",
            {"found": True, "data": b"\x00\x01\xff", "total": 3},
        )
        m2 = s.messages["FetchCheckpointResponse"]()
        m2.ParseFromString(m.SerializeToString())
        assert s.to_dict(m2)["data"] == b"\x00\x01\xff"


@pytest.mark.skipif(not os.path.isdir(REF_PROTOS),
                    reason="reference protos absent")
class TestReferenceWireCompat:
    """Bytes serialized by classes built from the REFERENCE's .proto
    files parse into ours with identical content, and vice versa — the
    wire format, not just the schema text, is compatible."""

    @pytest.fixture(scope="class")
    def ref(self):
        return Schema(REF_PROTOS, "reference")

    def test_done_request_ours_to_reference(self, ref):
        s = our_schema()
        b = s.to_message("DoneRequest", {
            "worker_id": 3, "job_id": [1, 2], "num_steps": [10, 20],
            "execution_time": [1.5, 2.5], "iterator_log": ["a", "b"],
        }).SerializeToString()
        m = ref.messages["DoneRequest"]()
        m.ParseFromString(b)
        d = ref.to_dict(m)
        assert d["job_id"] == [1, 2]
        assert d["execution_time"] == [1.5, 2.5]
        assert d["iterator_log"] == ["a", "b"]

    def test_lease_response_reference_to_ours(self, ref):
        b = ref.to_message("UpdateLeaseResponse", {
            "max_steps": 500, "max_duration": 120.5, "extra_time": 7.25,
            "run_time_so_far": 33, "deadline": 9000,
        }).SerializeToString()
        s = our_schema()
        m = s.messages["UpdateLeaseResponse"]()
        m.ParseFromString(b)
        d = s.to_dict(m)
        assert d["max_steps"] == 500
        assert d["max_duration"] == 120.5
        assert d["extra_time"] == 7.25
        assert d["run_time_so_far"] == 33 and d["deadline"] == 9000

    def test_heartbeat_nested_enum(self, ref):
        b = ref.to_message("Heartbeat", {
            "worker_id": 7,
            "job_state": [{"job_id": 4, "status": 2}],
        }).SerializeToString()
        s = our_schema()
        m = s.messages["Heartbeat"]()
        m.ParseFromString(b)
        d = s.to_dict(m)
        assert d["job_state"][0] == {"job_id": 4, "status": 2}
        # enum numbering aligned with the reference's enums.proto
        assert s.enums["JobStatus"]["RUNNING"] == 2
        assert s.enums["JobStatus"]["SUCCEEDED"] == 3

    def test_every_shared_message_field_for_field(self, ref):
        """Each message both schemas define carries the same
        (name, number, repeated) triples for the fields the REFERENCE
        declares (ours may extend into unused field space)."""
        s = our_schema()
        shared = set(s.messages) & set(ref.messages)
        assert len(shared) >= 13
        for name in shared:
            ours = {
                (f.name, f.number, f.is_repeated)
                for f in s.messages[name].DESCRIPTOR.fields
            }
            theirs = {
                (f.name, f.number, f.is_repeated)
                for f in ref.messages[name].DESCRIPTOR.fields
            }
            assert theirs <= ours, (
                f"{name}: reference fields {theirs - ours} missing here"
            )


class TestProtoCodecRpc:
    def test_scheduler_rpcs_over_proto_codec(self, monkeypatch):
        """The live gRPC stack end-to-end with protobuf bodies."""
        import socket

        monkeypatch.setenv("SWQ_RPC_CODEC", "proto")
        from shockwave_amd.rpc.services import (
            IteratorRpcClient, WorkerRpcClient, serve_scheduler,
        )

        sock = socket.socket()
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
        sock.close()

        seen = {}

        def register_worker(worker_type, num_gpus, ip_addr, port):
            seen["register"] = (worker_type, num_gpus, ip_addr, port)
            return [11, 12], 120

        def done(job_id, worker_id, all_num_steps, all_execution_times,
                 all_iterator_logs=None):
            seen["done"] = (job_id, worker_id, list(all_num_steps),
                            list(all_execution_times))

        def init_job(job_id):
            return 400, 300.0, 5.0, 17, 9000

        def update_lease(job_id, worker_id, steps, duration, max_steps,
                         max_duration):
            return 800, 600.0, 17, 9000

        server = serve_scheduler(port, {
            "RegisterWorker": register_worker,
            "Done": done,
            "InitJob": init_job,
            "UpdateLease": update_lease,
            "SendHeartbeat": lambda *a, **k: None,
            "UpdateResourceRequirement": lambda *a, **k: None,
        })
        try:
            wc = WorkerRpcClient("mi355x", "127.0.0.1", 50061,
                                 "127.0.0.1", port)
            worker_ids, round_duration, err = wc.register_worker(num_gpus=4)
            assert err is None
            assert list(worker_ids) == [11, 12] and round_duration == 120
            assert seen["register"][0] == "mi355x"

            ic = IteratorRpcClient(5, 11, "127.0.0.1", port)
            ms, md, extra, rt, dl = ic.init()
            assert (ms, md, extra, rt, dl) == (400, 300.0, 5.0, 17, 9000)
            ms, md = ic.update_lease(10, 1.5, 100, 60.0)[:2]
            assert ms == 800 and md == 600.0

            wc.notify_scheduler(11, [(5, 10, 1.5, "log line")])
            assert seen["done"][1] == 11
        finally:
            server.stop(0)
