#!/usr/bin/env python3
"""Single-frame service round-trip latency: anomalous LogSchema in ->
DetectorSchema alert back on the SAME channel (request/reply mode, no
out_addr). Measures the full deployment path: transport + packed decode
+ GPU pipeline + alert build + reply."""
import argparse
import json
import os
import subprocess
import sys
import tempfile
import time
import uuid

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import yaml  # noqa: E402

from detectmateservice_amd.engine.sockets import (  # noqa: E402
    PairDialer, RecvTimeout, ShmDialer,
)
from detectmateservice_amd.schemas import LogSchema  # noqa: E402
from detectmateservice_amd.utils.synthetic import (  # noqa: E402
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=2000)
    ap.add_argument("--warmup", type=int, default=200)
    ap.add_argument("--train-lines", type=int, default=5000)
    ap.add_argument("--shm", action="store_true")
    ap.add_argument("--no-packed", action="store_true")
    args = ap.parse_args()

    tmp = tempfile.mkdtemp(prefix="dmx-rtt-")
    uid = uuid.uuid4().hex[:6]
    addr = (f"shm:///dmx-rtt-{uid}" if args.shm
            else f"ipc://{tmp}/rtt-{uid}.ipc")
    tpl = os.path.join(tmp, "templates.txt")
    with open(tpl, "w") as fh:
        fh.write("\n".join(AUDIT_TEMPLATES) + "\n")

    def wy(name, data):
        p = os.path.join(tmp, name)
        with open(p, "w") as fh:
            yaml.safe_dump(data, fh)
        return p

    settings = wy("s.yaml", {
        "component_type": "FusedPipelineDetector",
        "engine_addr": addr,
        "out_addr": [],                       # request/reply mode
        "http_enabled": False,
        "engine_batch_size": 8192,
        "engine_batch_linger_ms": 0.0,
        "engine_recv_timeout": 100,
        "engine_packed_mode": not args.no_packed,
        "config_file": wy("c.yaml", {"detectors": {"FusedPipelineDetector": {
            "log_format": AUDIT_LOG_FORMAT,
            "path_templates": tpl,
            "watches": [{"kind": "header", "pos": 0}],
            "use_transformer": True,
            "score_threshold": 1.0e9,
            "data_use_training": args.train_lines,
        }}}),
        "log_dir": os.path.join(tmp, "logs"),
    })
    proc = subprocess.Popen(
        [sys.executable, "-m", "detectmateservice_amd.cli", "--settings",
         settings],
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    cli = (ShmDialer(addr) if args.shm else PairDialer(addr))
    try:
        assert cli.wait_connected(180.0)
        gen = AuditLogGenerator(seed=7, anomaly_rate=0.0)
        train = [LogSchema(logID=f"t{i}", log=gen.line()[0]).serialize()
                 for i in range(args.train_lines)]
        sent = 0
        while sent < len(train):
            n = cli.send_many(train[sent:sent + 4096], block=True)
            sent += n
        time.sleep(2.0)

        bad = ("type=ZZZ_RTT msg=audit(1.0:1): pid=1 uid=0 auid=1 ses=1 "
               "msg='op=PAM:x acct=\"x\" exe=/bin/x hostname=? addr=? "
               "terminal=x res=success'")
        lat = []
        for k in range(args.warmup + args.iters):
            frame = LogSchema(logID=f"r{k}", log=bad).serialize()
            t0 = time.perf_counter()
            while not cli.send(frame, block=False):
                time.sleep(0.0002)
            while True:
                try:
                    cli.recv(timeout_ms=10000)
                    break
                except RecvTimeout:
                    raise SystemExit("alert never came back")
            if k >= args.warmup:
                lat.append(time.perf_counter() - t0)
        lat.sort()
        p50 = lat[len(lat) // 2] * 1e3
        p99 = lat[int(len(lat) * 0.99)] * 1e3
        print(json.dumps({
            "metric": "service_rtt_ms",
            "value": round(p50, 3),
            "unit": "ms (p50)",
            "n_gpus": 1,
            "higher_is_better": False,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "transport": "shm" if args.shm else "ipc",
                "packed": not args.no_packed,
                "p99_ms": round(p99, 3),
                "iters": args.iters,
            },
        }))
    finally:
        cli.close()
        proc.terminate()
        try:
            proc.wait(timeout=8)
        except subprocess.TimeoutExpired:
            proc.kill()


if __name__ == "__main__":
    main()
