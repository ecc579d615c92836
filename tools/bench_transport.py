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
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 400_000
    tmp = tempfile.mkdtemp(prefix="dmx-tr-")
    addr = f"ipc://{tmp}/t-{uuid.uuid4().hex[:6]}.ipc"
    gen = AuditLogGenerator(seed=7, anomaly_rate=0.0)
    frames = [LogSchema(logID=f"m{i}", log=gen.line()[0]).serialize()
              for i in range(n)]
    total_bytes = sum(len(f) for f in frames)

    listener = PairListener(addr, buffer_size=8192)
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
    feeder = PairDialer(addr, buffer_size=8192)
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
    print(f"transport only: {n/dt:,.0f} lines/s  "
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


if __name__ == "__main__":
    main()
