#!/usr/bin/env python3
"""Isolate the service-mode transport ceiling: feeder -> pure drain sink
over the SP-framed ipc socket (no codec, no pipeline, no service).
Also times the C++ batched codec and the frame reader alone."""
import os
import sys
import tempfile
import threading
import time
import uuid

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from detectmateservice_amd.engine.sockets import PairDialer, PairListener, RecvTimeout
from detectmateservice_amd.schemas import LogSchema
from detectmateservice_amd.utils.synthetic import AuditLogGenerator


def main():
    # usage: bench_transport.py [lines] [scheme]   scheme in ipc|tcp|tls|ws
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 400_000
    scheme = sys.argv[2] if len(sys.argv) > 2 else "ipc"
    tmp = tempfile.mkdtemp(prefix="dmx-tr-")
    tls_in = tls_out = None
    if scheme == "ipc":
        addr = f"ipc://{tmp}/t-{uuid.uuid4().hex[:6]}.ipc"
    elif scheme in ("tcp", "ws"):
        addr = f"{scheme}://127.0.0.1:0"
    elif scheme == "tls":
        import subprocess

        from detectmateservice_amd.settings import TlsInputConfig, TlsOutputConfig

        key, crt, pem = f"{tmp}/key.pem", f"{tmp}/crt.pem", f"{tmp}/ck.pem"
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
             "-keyout", key, "-out", crt, "-days", "1", "-subj",
             "/CN=localhost", "-addext", "subjectAltName=DNS:localhost"],
            check=True, capture_output=True)
        with open(pem, "w") as fh:
            fh.write(open(key).read() + open(crt).read())
        tls_in = TlsInputConfig(cert_key_file=pem)
        tls_out = TlsOutputConfig(ca_file=crt, server_name="localhost")
        addr = "tls+tcp://127.0.0.1:0"
    else:
        raise SystemExit(f"unknown scheme {scheme}")
    gen = AuditLogGenerator(seed=7, anomaly_rate=0.0)
    frames = [LogSchema(logID=f"m{i}", log=gen.line()[0]).serialize()
              for i in range(n)]
    total_bytes = sum(len(f) for f in frames)

    listener = PairListener(addr, buffer_size=8192, tls_config=tls_in)
    if scheme in ("tcp", "ws", "tls"):
        addr = addr.replace(":0", f":{listener.bound_port}")
    got = [0]
    done = threading.Event()

    def drain():
        while got[0] < n:
            try:
                batch = listener.recv_many(max_frames=8192, timeout_ms=5000,
                                           linger_ms=1.0)
            except RecvTimeout:
                break
            got[0] += len(batch)
        done.set()

    t = threading.Thread(target=drain, daemon=True)
    t.start()
    feeder = PairDialer(addr, buffer_size=8192, tls_config=tls_out)
    assert feeder.wait_connected(10.0)
    t0 = time.perf_counter()
    sent = 0
    while sent < n:
        k = feeder.send_many(frames[sent:sent + 4096], block=False)
        if k == 0:
            time.sleep(0.0002)
        sent += k
    done.wait(30.0)
    dt = time.perf_counter() - t0
    print(f"[{scheme}] transport only: {n/dt:,.0f} lines/s  "
          f"({total_bytes/dt/1e6:.0f} MB/s)  recv={got[0]}")
    feeder.close()
    listener.close()

    # C++ codec alone
    from detectmateservice_amd import ops
    if ops.have_extension():
        from detectmateservice_amd.ops import _dmx_C
        batch = frames[:8192]
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            _dmx_C.decode_log_batch(batch, 256)
        dt = (time.perf_counter() - t0) / iters
        print(f"decode_log_batch(8192): {dt*1e3:.2f} ms "
              f"({8192/dt:,.0f} lines/s)")
        # the TLS/ws packed-mode fallback converts byte frames per batch
        t0 = time.perf_counter()
        for _ in range(iters):
            _dmx_C.decode_log_batch_packed(batch, 256, False)
        dt = (time.perf_counter() - t0) / iters
        print(f"decode_log_batch_packed(8192) [packed-fallback conversion]: "
              f"{dt*1e3:.2f} ms ({8192/dt:,.0f} lines/s)")


if __name__ == "__main__":
    main()
