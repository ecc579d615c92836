"""TLS transport (self-signed certs via openssl, reference
test_tls_transport.py shape), the injectable socket-factory seam, and the
OutputAggregator edge component."""
import subprocess
import threading
import time
from typing import List, Optional

import pytest

from detectmateservice_amd.engine.engine import Engine
from detectmateservice_amd.engine.sockets import (
    PairDialer,
    PairListener,
    RecvTimeout,
)
from detectmateservice_amd.settings import (
    ServiceSettings,
    TlsInputConfig,
    TlsOutputConfig,
)


@pytest.fixture(scope="module")
def certs(tmp_path_factory):
    """Self-signed server cert via openssl (reference shells out the same
    way, test_tls_transport.py:52-99)."""
    d = tmp_path_factory.mktemp("certs")
    key, crt, pem = d / "key.pem", d / "crt.pem", d / "cert_key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(crt), "-days", "1",
         "-subj", "/CN=localhost"],
        check=True, capture_output=True,
    )
    pem.write_text(crt.read_text() + key.read_text())
    return {"cert_key": pem, "ca": crt}


def test_tls_tcp_roundtrip(certs):
    listener = PairListener(
        "tls+tcp://127.0.0.1:0",
        tls_config=TlsInputConfig(cert_key_file=certs["cert_key"]),
    )
    port = listener.bound_port
    dialer = PairDialer(
        f"tls+tcp://127.0.0.1:{port}",
        tls_config=TlsOutputConfig(ca_file=certs["ca"], server_name="localhost"),
    )
    try:
        assert dialer.wait_connected(10.0)
        dialer.send(b"secret payload")
        assert listener.recv(timeout_ms=5000) == b"secret payload"
        listener.send(b"reply")
        assert dialer.recv(timeout_ms=5000) == b"reply"
    finally:
        dialer.close()
        listener.close()


def test_tls_engine_end_to_end(certs, tmp_path):
    """An engine listening on tls+tcp, a TLS client round trip."""
    settings = ServiceSettings(
        component_type="core",
        engine_addr="tls+tcp://127.0.0.1:0",
        tls_input=TlsInputConfig(cert_key_file=certs["cert_key"]),
        http_enabled=False,
        engine_recv_timeout=50,
    )

    class Echo:
        def process_batch(self, frames):
            return [b"tls:" + f for f in frames]

    eng = Engine(settings, Echo())
    port = eng._pair_sock.bound_port
    eng.start()
    client = PairDialer(
        f"tls+tcp://127.0.0.1:{port}",
        tls_config=TlsOutputConfig(ca_file=certs["ca"], server_name="localhost"),
    )
    try:
        assert client.wait_connected(10.0)
        client.send(b"hello")
        assert client.recv(timeout_ms=5000) == b"tls:hello"
    finally:
        client.close()
        eng.stop()
        eng.close()


# ---------------------------------------------------------------------------
# injectable socket-factory seam (reference engine.py:111-113)
# ---------------------------------------------------------------------------


class FakeSocket:
    def __init__(self):
        self.inbox: List[bytes] = []
        self.sent: List[bytes] = []
        self.closed = False

    def recv(self, timeout_ms=None):
        if self.inbox:
            return self.inbox.pop(0)
        raise RecvTimeout("fake")

    def recv_many(self, max_frames, timeout_ms, linger_ms=0.0):
        out, self.inbox = self.inbox[:max_frames], self.inbox[max_frames:]
        if not out:
            time.sleep(timeout_ms / 1000.0)
        return out

    def send(self, data, block=True):
        self.sent.append(data)
        return True

    def close(self):
        self.closed = True


class FakeFactory:
    def __init__(self):
        self.listener = FakeSocket()
        self.dialers = {}

    def create(self, addr, logger=None, tls_config=None, buffer_size=128):
        return self.listener

    def create_dialer(self, addr, logger=None, tls_config=None, buffer_size=128):
        d = FakeSocket()
        d.addr = addr
        self.dialers[addr] = d
        return d


def test_engine_with_fake_transport(tmp_path):
    factory = FakeFactory()
    settings = ServiceSettings(
        component_type="core",
        engine_addr="inproc://fake-seam",
        out_addr=["inproc://fake-out-a", "inproc://fake-out-b"],
        http_enabled=False,
        engine_recv_timeout=20,
    )

    class Upper:
        def process_batch(self, frames):
            return [f.upper() for f in frames]

    eng = Engine(settings, Upper(), socket_factory=factory)
    factory.listener.inbox = [b"one", b"two"]
    eng.start()
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        if all(len(d.sent) >= 2 for d in factory.dialers.values()):
            break
        time.sleep(0.02)
    eng.stop()
    eng.close()
    for d in factory.dialers.values():
        assert d.sent == [b"ONE", b"TWO"]
    assert factory.listener.closed


# ---------------------------------------------------------------------------
# OutputAggregator
# ---------------------------------------------------------------------------


def test_output_aggregator_windows(tmp_path):
    from detectmateservice_amd.library.outputs import OutputAggregator
    from detectmateservice_amd.schemas import DetectorSchema, OutputSchema

    out_file = tmp_path / "alerts.jsonl"
    agg = OutputAggregator({"window_size": 2, "output_file": str(out_file)})
    a1 = DetectorSchema(detectorID="d1", detectorType="nvd", alertID="a1",
                        logIDs=["l1"], description="first",
                        alertsObtain={"Global - URL": "/x"}).serialize()
    a2 = DetectorSchema(detectorID="d2", detectorType="tf", alertID="a2",
                        logIDs=["l2"], description="second").serialize()
    assert agg.process(a1) is None
    out = agg.process(a2)
    assert out is not None
    o = OutputSchema.deserialize(out)
    assert o.detectorIDs == ["d1", "d2"]
    assert o.alertIDs == ["a1", "a2"]
    assert o.logIDs == ["l1", "l2"]
    assert "first" in o.description and "second" in o.description
    assert o.alertsObtain["Global - URL"] == "/x"
    assert out_file.exists() and "a1" in out_file.read_text()


def test_output_aggregator_resolvable():
    from detectmateservice_amd.components.resolver import ComponentResolver

    path, _ = ComponentResolver().resolve("OutputAggregator")
    assert path.endswith("OutputAggregator")


def test_ipc_unlink_failure_is_tolerated(tmp_path, monkeypatch):
    """OS error while unlinking a stale ipc socket file is logged, not
    fatal (reference test_engine_socket_factory_error_handling.py:74-83)."""
    from pathlib import Path as _P

    addr = f"ipc://{tmp_path}/stale.ipc"
    (tmp_path / "stale.ipc").write_bytes(b"")  # stale file

    real_unlink = _P.unlink

    def bad_unlink(self, *a, **kw):
        if self.name == "stale.ipc":
            raise OSError("simulated unlink failure")
        return real_unlink(self, *a, **kw)

    monkeypatch.setattr(_P, "unlink", bad_unlink)
    # bind then fails (address in use by the stale file) OR succeeds on
    # platforms that allow rebinding — either way no crash beyond OSError
    try:
        l = PairListener(addr)
        l.close()
    except OSError:
        pass


def test_engine_setup_failure_closes_input(ipc_addr):
    """Output-socket setup failure must close the bound input socket
    (reference engine.py:122-129)."""
    from detectmateservice_amd.engine.engine import Engine
    from detectmateservice_amd.settings import ServiceSettings

    class ExplodingFactory:
        def __init__(self):
            self.listener = None

        def create(self, addr, logger=None, tls_config=None, buffer_size=128):
            from detectmateservice_amd.engine.sockets import PairListener

            self.listener = PairListener(addr, buffer_size=buffer_size)
            return self.listener

        def create_dialer(self, *a, **kw):
            raise RuntimeError("dialer setup boom")

    factory = ExplodingFactory()
    settings = ServiceSettings(
        component_type="core", engine_addr=ipc_addr,
        out_addr=["ipc:///tmp/never-exists-xyz.ipc"], http_enabled=False,
    )
    with pytest.raises(RuntimeError):
        Engine(settings, processor=None, socket_factory=factory)
    assert factory.listener is not None
    assert factory.listener._closed.is_set()
