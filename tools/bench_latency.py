#!/usr/bin/env python3
"""Latency-mode characterization: per-batch detect latency across batch
sizes, eager vs hipGraph replay (the serving-latency complement to
bench.py's throughput mode)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from detectmateservice_amd import ops  # noqa: E402
from detectmateservice_amd.pipeline import GpuPipeline, PipelineConfig  # noqa: E402
from detectmateservice_amd.utils.synthetic import (  # noqa: E402
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)


def main():
    device = "cuda" if torch.cuda.is_available() else "cpu"
    gen = AuditLogGenerator(seed=3, anomaly_rate=0.01)
    results = []
    for B in (512, 2048, 8192):
        cfg = PipelineConfig(
            templates=AUDIT_TEMPLATES, log_format=AUDIT_LOG_FORMAT,
            watches=[{"kind": "variable", "pos": 5, "event": 1}],
            train_lines=B, use_transformer=True, max_len=256,
        )
        pipe = GpuPipeline(cfg, device=device)
        pool = []
        for _ in range(4):
            raw = [gen.line()[0].encode() for _ in range(B)]
            l, n = ops.pack_lines(raw, 256, device=device)
            pool.append((l, n))
        pipe.process_packed(*pool[0])  # train
        for mode in ("eager", "graph"):
            if mode == "graph":
                if device != "cuda":
                    continue
                pipe.enable_graph(B)
                run = pipe.process_packed_graph
            else:
                run = pipe.process_packed
            for i in range(5):
                run(*pool[i % 4])
            if device == "cuda":
                torch.cuda.synchronize()
            times = []
            for i in range(30):
                t0 = time.perf_counter()
                run(*pool[i % 4])
                if device == "cuda":
                    torch.cuda.synchronize()
                times.append((time.perf_counter() - t0) * 1000)
            times.sort()
            results.append({
                "batch": B, "mode": mode,
                "p50_ms": round(times[len(times) // 2], 3),
                "p99_ms": round(times[-1], 3),
                "lines_per_sec": round(B / (times[len(times) // 2] / 1000), 0),
            })
    print(json.dumps(results, indent=2))


if __name__ == "__main__":
    main()
