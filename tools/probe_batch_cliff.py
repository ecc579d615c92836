"""Single-launch batch-size sweep of Pipeline.process_packed.

The service at engine batch >=32768 collapsed to ~1M lines/s while
bench.py at 65536 runs 7M — but bench auto-shards into 16384-line
chunks, so one-launch batches >=32768 were never timed in isolation.
This probe times process_packed at one B per launch to locate the
cliff; pair with `rocprofv3 --stats` to name the kernel responsible.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from detectmateservice_amd.pipeline import GpuPipeline, PipelineConfig
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)
from detectmateservice_amd import ops


def main():
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = PipelineConfig(
        templates=list(AUDIT_TEMPLATES),
        log_format=AUDIT_LOG_FORMAT,
        watches=[{"kind": "variable", "pos": 5, "event": 1}],
        train_lines=0,
        use_transformer=True,
        score_threshold=1.0e9,
        max_len=256,
    )
    pipe = GpuPipeline(cfg, device=dev)
    gen = AuditLogGenerator(seed=7, anomaly_rate=0.0)
    maxB = 65536
    raw = [gen.line()[0].encode() for _ in range(8192)]
    lines_cpu, lens_cpu = ops.pack_lines(
        [raw[i % len(raw)] for i in range(maxB)], cfg.max_len)
    lines = lines_cpu.to(dev)
    lens = lens_cpu.to(dev)

    sizes = [int(x) for x in (sys.argv[1].split(",") if len(sys.argv) > 1 else
             "8192,16384,24576,28672,30720,32768,36864,49152,65536".split(","))]
    for B in sizes:
        l, n = lines[:B], lens[:B]
        for _ in range(3):
            pipe.process_packed(l, n)
        torch.cuda.synchronize() if torch.cuda.is_available() else None
        t0 = time.perf_counter()
        iters = 10
        for _ in range(iters):
            pipe.process_packed(l, n)
        torch.cuda.synchronize() if torch.cuda.is_available() else None
        dt = (time.perf_counter() - t0) / iters
        print(f"B={B:6d}  {dt*1e3:8.3f} ms/launch  {B/dt/1e6:7.2f} M lines/s",
              flush=True)


if __name__ == "__main__":
    main()
