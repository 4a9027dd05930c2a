"""Unit tests for L0 utilities (serialization round-trip,
nested pack/unpack, MPFuture pair semantics, TimedStorage expiration)."""
import threading
import time
from concurrent.futures import TimeoutError as FutureTimeoutError

import pytest
import torch

from learning_at_home_amd.utils import (
    MPFuture,
    MSGPackSerializer,
    PytorchSerializer,
    TimedStorage,
    get_dht_time,
    nested_compare,
    nested_flatten,
    nested_map,
    nested_pack,
)


class TestSerializer:
    @pytest.mark.parametrize("dtype", [torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.bool])
    def test_tensor_roundtrip(self, dtype):
        t = (torch.randn(4, 5) * 10).to(dtype)
        out = PytorchSerializer.loads(PytorchSerializer.dumps(t))
        assert out.dtype == dtype and out.shape == t.shape
        assert torch.equal(out.float(), t.float())

    def test_requires_grad_preserved(self):
        t = torch.randn(3, requires_grad=True)
        out = PytorchSerializer.loads(PytorchSerializer.dumps(t))
        assert out.requires_grad

    def test_nested_structures(self):
        obj = {"a": [torch.ones(2), {"b": torch.zeros(1, 3)}], "c": "hello", "d": 3.5}
        out = PytorchSerializer.loads(PytorchSerializer.dumps(obj))
        assert torch.equal(out["a"][0], torch.ones(2))
        assert torch.equal(out["a"][1]["b"], torch.zeros(1, 3))
        assert out["c"] == "hello" and out["d"] == 3.5

    def test_msgpack_plain(self):
        obj = {"k": [1, 2, "three"], "n": None}
        assert MSGPackSerializer.loads(MSGPackSerializer.dumps(obj)) == obj

    def test_empty_tensor(self):
        t = torch.empty(0, 5)
        out = PytorchSerializer.loads(PytorchSerializer.dumps(t))
        assert out.shape == (0, 5)


class TestNested:
    def test_flatten_pack_roundtrip(self):
        s = {"x": (1, [2, 3]), "y": {"z": 4}}
        flat = list(nested_flatten(s))
        assert flat == [1, 2, 3, 4]
        assert nested_pack(flat, s) == s

    def test_map(self):
        s = (1, {"a": 2})
        assert nested_map(lambda v: v * 2, s) == (2, {"a": 4})

    def test_map_multi(self):
        assert nested_map(lambda a, b: a + b, (1, 2), (10, 20)) == (11, 22)

    def test_compare(self):
        assert nested_compare({"a": (1, 2)}, {"a": (9, 9)})
        assert not nested_compare({"a": (1, 2)}, {"a": (1, 2, 3)})

    def test_pack_wrong_arity(self):
        with pytest.raises(ValueError):
            nested_pack([1, 2, 3], (1, 2))


class TestMPFuture:
    def test_inprocess(self):
        f = MPFuture()
        threading.Thread(target=lambda: f.set_result(42)).start()
        assert f.result(timeout=2.0) == 42
        assert f.done()

    def test_exception(self):
        f = MPFuture()
        f.set_exception(ValueError("boom"))
        with pytest.raises(ValueError):
            f.result(0.1)

    def test_timeout(self):
        f = MPFuture()
        with pytest.raises(FutureTimeoutError):
            f.result(timeout=0.05)

    def test_cancel(self):
        f = MPFuture()
        assert f.cancel()
        assert f.cancelled()

    def test_pair_across_process(self):
        import multiprocessing as mp

        setter, awaiter = MPFuture.make_pair()

        def child(s):
            s.set_result(torch.ones(3))

        p = mp.Process(target=child, args=(setter,))
        p.start()
        result = awaiter.result(timeout=10.0)
        assert torch.equal(result, torch.ones(3))
        p.join()


class TestTimedStorage:
    def test_store_get(self):
        s = TimedStorage()
        assert s.store("k", "v", get_dht_time() + 10)
        value, exp = s.get("k")
        assert value == "v" and exp > get_dht_time()

    def test_expiration(self):
        s = TimedStorage()
        s.store("k", "v", get_dht_time() + 0.1)
        time.sleep(0.15)
        assert s.get("k") == (None, None)

    def test_fresher_wins(self):
        s = TimedStorage()
        t0 = get_dht_time()
        s.store("k", "old", t0 + 10)
        assert not s.store("k", "older", t0 + 5)  # staler value rejected
        assert s.store("k", "new", t0 + 20)
        assert s.get("k")[0] == "new"

    def test_maxsize_eviction(self):
        s = TimedStorage(maxsize=2)
        t0 = get_dht_time()
        s.store("a", 1, t0 + 5)
        s.store("b", 2, t0 + 10)
        s.store("c", 3, t0 + 15)
        assert len(s) == 2
        assert s.get("a") == (None, None)  # soonest-to-expire evicted


class TestMonitor:
    def test_stage_timers_cpu(self):
        from learning_at_home_amd.utils.monitor import StageTimers

        t = StageTimers(use_gpu_events=False)
        with t("work"):
            sum(range(1000))
        with t("work"):
            pass
        s = t.summary()
        assert s["work"]["count"] == 2 and s["work"]["total_s"] >= 0

    def test_expert_load_stats(self):
        import torch
        from learning_at_home_amd.utils.monitor import expert_load_stats

        stats = expert_load_stats(torch.tensor([10, 10, 10, 10]), capacity=8)
        assert stats["mean_load"] == 10.0 and stats["cv"] == 0.0
        assert stats["dropped_frac"] == 0.2  # 2 over capacity per expert
        assert stats["entropy_frac"] == 1.0


class TestConnectionFraming:
    def test_sync_roundtrip(self):
        import socket
        import threading
        from learning_at_home_amd.utils import recv_message, send_message

        server = socket.socket()
        server.bind(("127.0.0.1", 0))
        server.listen(1)
        port = server.getsockname()[1]
        got = {}

        def serve():
            conn, _ = server.accept()
            got["msg"] = recv_message(conn)
            send_message(conn, "pong", b"\x00" * 100000)  # multi-chunk reply
            conn.close()

        t = threading.Thread(target=serve)
        t.start()
        client = socket.create_connection(("127.0.0.1", port))
        send_message(client, "ping", b"payload")
        kind, payload = recv_message(client)
        t.join(5)
        assert got["msg"] == ("ping", b"payload")
        assert kind == "pong" and len(payload) == 100000
        client.close()
        server.close()

    def test_serializer_pickle_fallback_gated(self):
        # complex(...) is not msgpack-encodable -> the torch.save path requires
        # explicit allow_pickle on BOTH ends (trusted same-host pipes only)
        import pytest

        from learning_at_home_amd.utils import PytorchSerializer

        obj = {"z": complex(1, 2)}
        with pytest.raises(TypeError):
            PytorchSerializer.dumps(obj)  # network-path default: no pickle out
        blob = PytorchSerializer.dumps(obj, allow_pickle=True)
        with pytest.raises(ValueError):
            PytorchSerializer.loads(blob)  # network-path default: no pickle in
        assert PytorchSerializer.loads(blob, allow_pickle=True) == obj

    def test_serializer_exception_roundtrip_no_pickle(self):
        # exceptions ride msgpack (type name + message), never pickle
        from learning_at_home_amd.utils import PytorchSerializer
        from learning_at_home_amd.utils.serializer import RemoteException

        blob = PytorchSerializer.dumps(["exception", ValueError("boom")])
        assert blob[:1] == b"M"
        kind, exc = PytorchSerializer.loads(blob)
        assert isinstance(exc, ValueError) and str(exc) == "boom"

        class CustomError(Exception):
            pass

        blob = PytorchSerializer.dumps(CustomError("custom boom"))
        exc = PytorchSerializer.loads(blob)
        assert isinstance(exc, RemoteException)
        assert exc.type_name == "CustomError" and "custom boom" in str(exc)

    def test_malformed_dht_datagram_ignored(self):
        from learning_at_home_amd.dht.protocol import DHTProtocol
        from learning_at_home_amd.dht.routing import DHTID

        proto = DHTProtocol(DHTID.generate())
        proto.datagram_received(b"\xff\x00garbage", ("127.0.0.1", 1))  # must not raise


class TestMonitorExtra:
    def test_stage_timers_cpu(self):
        import time as _time

        from learning_at_home_amd.utils.monitor import StageTimers

        timers = StageTimers(use_gpu_events=False)
        for _ in range(3):
            with timers("work"):
                _time.sleep(0.01)
        timers.flush()
        s = timers.summary()
        assert s["work"]["count"] == 3
        assert s["work"]["total_s"] >= 0.03
        assert "work" in timers.report()

    def test_expert_load_stats_uniform_and_skewed(self):
        from learning_at_home_amd.utils.monitor import expert_load_stats

        uniform = expert_load_stats(torch.full((8,), 100.0), capacity=128)
        assert uniform["dropped_frac"] == 0.0
        assert uniform["cv"] == 0.0
        assert abs(uniform["entropy_frac"] - 1.0) < 1e-6
        skewed = expert_load_stats(torch.tensor([800.0] + [0.0] * 7), capacity=128)
        assert skewed["max_load"] == 800 and skewed["min_load"] == 0
        assert skewed["dropped_frac"] == round((800 - 128) / 800, 4)
        assert skewed["entropy_frac"] < 0.01  # all mass on one expert


class TestFramingGuards:
    def test_oversized_frame_rejected(self):
        """A length prefix beyond MAX_MESSAGE_BYTES must raise, not allocate."""
        import socket

        from learning_at_home_amd.utils import recv_message
        from learning_at_home_amd.utils.connection import _LEN, MAX_MESSAGE_BYTES

        a, b = socket.socketpair()
        try:
            a.sendall(_LEN.pack(MAX_MESSAGE_BYTES + 1))
            b.settimeout(5.0)
            with pytest.raises(ValueError, match="too large"):
                recv_message(b)
        finally:
            a.close()
            b.close()


def test_tesseract_codename_aliases():
    """The reference's original class names (Tesseract*) must resolve to the
    same objects as the modern names — drop-in parity for old import paths."""
    import learning_at_home_amd as lah

    assert lah.TesseractServer is lah.Server
    assert lah.TesseractNetwork is lah.DHT
    assert lah.TesseractRuntime is lah.Runtime
