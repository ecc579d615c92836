"""Engine loop tests: echo, None-skip, error-survive, broadcast, drop
accounting, batch path (reference tests/test_engine_loop.py and
test_engine_multi_output.py shapes)."""
import time
from typing import List, Optional

import pytest

from detectmateservice_amd.engine.engine import Engine, EngineException
from detectmateservice_amd.engine.sockets import PairDialer, PairListener, RecvTimeout
from detectmateservice_amd.settings import ServiceSettings


class SimpleProcessor:
    """Echo with a prefix."""

    def process(self, data: bytes) -> Optional[bytes]:
        return b"out:" + data

    def process_batch(self, frames: List[bytes]) -> List[Optional[bytes]]:
        return [self.process(f) for f in frames]


class NullProcessor:
    def process(self, data: bytes) -> Optional[bytes]:
        return None

    def process_batch(self, frames):
        return [None] * len(frames)


class FailingProcessor:
    def process(self, data: bytes) -> Optional[bytes]:
        raise RuntimeError("boom")

    def process_batch(self, frames):
        raise RuntimeError("boom")


class BatchRecorder:
    """Records batch sizes, echoes frames."""

    def __init__(self):
        self.batches = []

    def process(self, data):
        return data

    def process_batch(self, frames):
        self.batches.append(len(frames))
        return list(frames)


def _settings(ipc_addr, out_addrs=(), **kw):
    defaults = dict(
        component_type="core",
        engine_addr=ipc_addr,
        out_addr=list(out_addrs),
        engine_recv_timeout=50,
        engine_retry_count=2,
        engine_batch_size=64,
        engine_batch_linger_ms=5.0,
        http_enabled=False,
    )
    defaults.update(kw)
    return ServiceSettings(**defaults)


@pytest.fixture
def run_engine(ipc_addr):
    engines = []

    def _make(processor, out_addrs=(), **kw):
        eng = Engine(_settings(ipc_addr, out_addrs, **kw), processor)
        eng.start()
        engines.append(eng)
        return eng

    yield _make
    for eng in engines:
        try:
            eng.stop()
        except EngineException:
            pass
        eng.close()


def test_echo_reply_mode(ipc_addr, run_engine):
    """No outputs → reply on the input socket (reference engine.py:248-264)."""
    run_engine(SimpleProcessor())
    client = PairDialer(ipc_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(b"ping")
        assert client.recv(timeout_ms=3000) == b"out:ping"
    finally:
        client.close()


def test_none_filters_message(ipc_addr, run_engine):
    """None result → nothing sent; downstream recv times out
    (reference test shape: engine_loop None-skip)."""
    run_engine(NullProcessor())
    client = PairDialer(ipc_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(b"ping")
        with pytest.raises(RecvTimeout):
            client.recv(timeout_ms=300)
    finally:
        client.close()


def test_processing_error_survives(ipc_addr, run_engine):
    """process() raising keeps the loop alive and counts errors
    (reference engine.py:233-236)."""
    eng = run_engine(FailingProcessor())
    client = PairDialer(ipc_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(b"a")
        client.send(b"b")
        time.sleep(0.3)
        assert eng.running
        with pytest.raises(RecvTimeout):
            client.recv(timeout_ms=200)
    finally:
        client.close()


def test_multi_output_broadcast(ipc_addr, tmp_path, run_engine):
    """Each processed frame goes to ALL outputs (reference
    test_engine_multi_output.py:139-155)."""
    outs = [f"ipc://{tmp_path}/out{i}.ipc" for i in range(3)]
    receivers = [PairListener(a) for a in outs]
    try:
        run_engine(SimpleProcessor(), out_addrs=outs)
        client = PairDialer(ipc_addr)
        try:
            assert client.wait_connected(5.0)
            client.send(b"msg")
            for r in receivers:
                assert r.recv(timeout_ms=5000) == b"out:msg"
        finally:
            client.close()
    finally:
        for r in receivers:
            r.close()


def test_partial_output_failure(ipc_addr, tmp_path, run_engine):
    """A dead output never stops delivery to the live ones (reference
    test_engine_multi_output.py:210-231); drops are counted."""
    live_addr = f"ipc://{tmp_path}/live.ipc"
    dead_addr = f"ipc://{tmp_path}/dead.ipc"
    live = PairListener(live_addr)
    try:
        eng = run_engine(
            SimpleProcessor(), out_addrs=[live_addr, dead_addr],
            engine_buffer_size=1, engine_retry_count=1,
        )
        client = PairDialer(ipc_addr)
        try:
            assert client.wait_connected(5.0)
            for i in range(5):
                client.send(b"m%d" % i)
            got = []
            deadline = time.monotonic() + 5
            while len(got) < 5 and time.monotonic() < deadline:
                try:
                    got.append(live.recv(timeout_ms=300))
                except RecvTimeout:
                    break
            # At buffer_size=1 individual frames may be dropped (drop-don't-
            # block), but delivery to the live output continues and order is
            # preserved for what arrives.
            assert len(got) >= 3
            sent = [b"out:m%d" % i for i in range(5)]
            it = iter(sent)
            assert all(any(g == s for s in it) for g in got), "order not preserved"
            # dead output (and any buffer-full races) accumulated drops
            deadline = time.monotonic() + 5
            dropped = 0
            while time.monotonic() < deadline:
                dropped = eng.metrics.data_dropped_lines_total._value.get()
                if dropped > 0:
                    break
                time.sleep(0.05)
            assert dropped > 0
        finally:
            client.close()
    finally:
        live.close()


def test_startup_with_absent_downstream(ipc_addr, tmp_path, run_engine):
    """Outputs may be unreachable at startup; late-bound listener gets
    buffered frames (reference test_engine_multi_output.py:371-409)."""
    out_addr = f"ipc://{tmp_path}/late.ipc"
    eng = run_engine(SimpleProcessor(), out_addrs=[out_addr], engine_buffer_size=64)
    assert eng.running
    client = PairDialer(ipc_addr)
    try:
        assert client.wait_connected(5.0)
        client.send(b"early")
        time.sleep(0.2)
        late = PairListener(out_addr)
        try:
            assert late.recv(timeout_ms=5000) == b"out:early"
        finally:
            late.close()
    finally:
        client.close()


def test_batched_processing(ipc_addr, run_engine):
    """A burst of frames reaches the processor as batches, order preserved."""
    rec = BatchRecorder()
    run_engine(rec, engine_batch_linger_ms=50.0)
    client = PairDialer(ipc_addr, buffer_size=256)
    try:
        assert client.wait_connected(5.0)
        n = 50
        for i in range(n):
            client.send(b"b%02d" % i)
        got = []
        deadline = time.monotonic() + 10
        while len(got) < n and time.monotonic() < deadline:
            try:
                got.append(client.recv(timeout_ms=300))
            except RecvTimeout:
                pass
        assert got == [b"b%02d" % i for i in range(n)]
        assert max(rec.batches) > 1  # actually batched
    finally:
        client.close()


def test_engine_restart(ipc_addr):
    """Engine thread is re-created on restart (reference engine.py:185-191)."""
    eng = Engine(_settings(ipc_addr), SimpleProcessor())
    try:
        eng.start()
        assert eng.running
        eng.stop()
        assert not eng.running
        eng.start()
        assert eng.running
        client = PairDialer(ipc_addr)
        try:
            assert client.wait_connected(5.0)
            client.send(b"again")
            assert client.recv(timeout_ms=3000) == b"out:again"
        finally:
            client.close()
    finally:
        eng.stop()
        eng.close()


def test_batch_size_one_strict_per_message(ipc_addr, run_engine):
    """engine_batch_size=1 degenerates to the reference's strict
    per-message loop with identical semantics."""
    run_engine(SimpleProcessor(), engine_batch_size=1, engine_batch_linger_ms=0.0)
    client = PairDialer(ipc_addr)
    try:
        assert client.wait_connected(5.0)
        for i in range(10):
            client.send(b"s%d" % i)
        got = []
        deadline = time.monotonic() + 10
        while len(got) < 10 and time.monotonic() < deadline:
            try:
                got.append(client.recv(timeout_ms=300))
            except RecvTimeout:
                pass
        assert got == [b"out:s%d" % i for i in range(10)]
    finally:
        client.close()


def test_packed_loop_survives_component_errors(tmp_path):
    """A component whose submit raises must not kill the packed loop;
    processing_errors_total counts and later batches still process
    (reference engine.py:233-236 error policy on the native plane)."""
    import time as time_mod

    from detectmateservice_amd import ops

    if not ops.have_extension():
        pytest.skip("extension not built")
    from detectmateservice_amd.engine.engine import Engine
    from detectmateservice_amd.engine.sockets import PairDialer
    from detectmateservice_amd.schemas import LogSchema
    from detectmateservice_amd.settings import ServiceSettings

    calls = {"n": 0}
    got = []

    class Flaky:
        def process(self, d):
            return None

        def process_batch(self, frames):
            return [None] * len(frames)

        def supports_packed_frames(self):
            return True

        def packed_max_len(self):
            return 64

        def packed_pin_memory(self):
            return False

        def process_packed_frames(self, lines, lens, blob, off):
            calls["n"] += 1
            if calls["n"] == 1:
                raise RuntimeError("boom")
            got.append(int(lines.shape[0]))
            return []

    addr = f"ipc://{tmp_path}/pkerr.ipc"
    settings = ServiceSettings(
        component_type="core", engine_addr=addr, http_enabled=False,
        engine_packed_mode=True, engine_batch_size=64,
        log_dir=tmp_path / "logs",
    )
    engine = Engine(settings, processor=Flaky())
    engine.start()
    dialer = PairDialer(addr)
    try:
        assert dialer.wait_connected(10.0)
        frames = [LogSchema(logID=str(i), log="x").serialize() for i in range(8)]
        for f in frames:
            assert dialer.send(f, block=True)
        deadline = time_mod.monotonic() + 10.0
        while calls["n"] < 1 and time_mod.monotonic() < deadline:
            time_mod.sleep(0.05)
        # second batch after the error must be processed
        for f in frames:
            assert dialer.send(f, block=True)
        while not got and time_mod.monotonic() < deadline:
            time_mod.sleep(0.05)
        assert calls["n"] >= 2 and got, (calls, got)
        assert engine.metrics.processing_errors_total._value.get() >= 8
    finally:
        dialer.close()
        engine.stop()
        engine.close()


class RecordingReplySock:
    """Fake pair socket recording reply(idx) calls (seam: reference
    engine_socket.py:23-32 factory injection)."""

    def __init__(self, batches):
        self.addr = "inproc://fake"
        self._batches = list(batches)
        self.replies = []

    def recv_many(self, max_frames, timeout_ms, linger_ms=0.0):
        if self._batches:
            return self._batches.pop(0)
        raise RecvTimeout(self.addr)

    def reply(self, idx, data):
        self.replies.append((idx, data))
        return True

    def send(self, data, block=True):
        self.replies.append((None, data))
        return True

    def close(self):
        pass


class RecordingFactory:
    def __init__(self, sock):
        self.sock = sock

    def create(self, addr, **kw):
        return self.sock

    def create_dialer(self, addr, **kw):
        raise AssertionError("no outputs expected")


def test_reply_indices_survive_empty_frame_filtering(ipc_addr):
    """Empty frames are skipped (reference engine.py:207-209) but reply
    routing must use each surviving frame's ORIGINAL batch index —
    _batch_conns was recorded for the unfiltered batch (ADVICE low #1)."""
    sock = RecordingReplySock([[b"", b"x", b"", b"y"]])
    eng = Engine(_settings(ipc_addr), SimpleProcessor(),
                 socket_factory=RecordingFactory(sock))
    eng.start()
    deadline = time.time() + 5.0
    while len(sock.replies) < 2 and time.time() < deadline:
        time.sleep(0.01)
    eng.stop()
    # frames at original indices 1 ("x") and 3 ("y") get the replies
    assert sock.replies == [(1, b"out:x"), (3, b"out:y")]
