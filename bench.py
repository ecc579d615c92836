#!/usr/bin/env python3
"""Flagship benchmark: lines/sec through reader→parser→detector.

Measures the BASELINE.json headline metric — synthetic audit-log lines
flowing through the fused GPU pipeline (template-match parser kernel →
NewValue hash-set probe → BERT-tiny bf16 MFMA transformer detector) — on
1..N MI355X GPUs, one rank per GPU, data-parallel detectors with an RCCL
all-gather of per-rank anomaly summaries each step (BASELINE config 4).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for
N>1 the driver launches via torch.distributed.run with one rank per GPU.
W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides; elapsed = MAX over ranks;
rank 0 prints ONE JSON line.

The synthetic line pool is generated once (seeded per rank) and resides in
HBM (the "GPU line buffer" design, BASELINE config 5); every step parses,
hashes, probes and scores a fresh slice — no cached outputs, all stages
execute inside the timed region.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import time

import torch

from detectmateservice_amd import ops
from detectmateservice_amd.pipeline import GpuPipeline, PipelineConfig
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=65536, help="lines per rank per step")
    p.add_argument("--pool", type=int, default=4, help="distinct pre-generated batches")
    p.add_argument("--max-len", type=int, default=256)
    p.add_argument("--no-transformer", action="store_true")
    p.add_argument("--graph", action="store_true", help="hipGraph-capture the steady-state step (pays at small/latency batches; neutral at 32k where the input copy offsets launch savings)")
    p.add_argument("--device", default=None)
    return p.parse_args()


def main() -> None:
    args = parse_args()
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
        if device.type == "cuda":
            torch.cuda.set_device(device)
    elif use_gpu:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        device = torch.device("cuda", local_rank % torch.cuda.device_count())
    else:
        device = torch.device("cpu")

    dist = None
    if world_size > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        import datetime

        dist.init_process_group(
            backend="nccl" if use_gpu else "gloo",
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(seconds=300),
        )

    # ---- build the pipeline (random-init weights, synthetic shapes) ----
    cfg = PipelineConfig(
        templates=AUDIT_TEMPLATES,
        log_format=AUDIT_LOG_FORMAT,
        watches=[
            {"kind": "variable", "pos": 5, "event": 1},   # acct
            {"kind": "variable", "pos": 6, "event": 1},   # exe
            {"kind": "header", "pos": 0, "event": -1},    # Type
        ],
        train_lines=args.batch,  # first (warmup) batch trains the hash sets
        use_transformer=not args.no_transformer,
        score_threshold=3.0,
        max_len=args.max_len,
        seed=1234,
    )
    pipe = GpuPipeline(cfg, device=device)

    # ---- synthetic line pool, resident on device ----
    gen = AuditLogGenerator(seed=1000 + rank, anomaly_rate=0.01)
    pool = []
    for _ in range(args.pool):
        raw = [gen.line()[0].encode() for _ in range(args.batch)]
        lines, lens = ops.pack_lines(raw, args.max_len, device="cpu")
        pool.append((lines.to(device), lens.to(device)))

    def sync():
        if use_gpu:
            torch.cuda.synchronize()

    def barrier():
        if dist is not None:
            dist.barrier()

    use_graph = [False]

    def step(i: int) -> None:
        lines, lens = pool[i % len(pool)]
        if use_graph[0]:
            out = pipe.process_packed_graph(lines, lens)
        else:
            out = pipe.process_packed(lines, lens)
        if dist is not None:
            # DP aggregation over RCCL/xGMI: per-rank anomaly summary
            summary = torch.stack(
                [out["anomaly"].sum().float(), out["scores"].sum()]
            ).to(device)
            gathered = [torch.empty_like(summary) for _ in range(world_size)]
            dist.all_gather(gathered, summary)

    # ---- warmup (also trains the NewValue hash sets on the first batch) ----
    for i in range(args.warmup):
        step(i)
    sync()
    # capture the steady-state detect path as ONE hipGraph (HIP graphs for
    # the launch-bound loop); fall back silently when capture is unsupported
    if use_gpu and args.graph:
        try:
            if pipe.enable_graph(args.batch):
                use_graph[0] = True
        except Exception as exc:  # noqa: BLE001
            if rank == 0:
                print(f"# graph capture disabled: {exc}", flush=True)
    sync()
    barrier()
    sync()

    # ---- timed region: EXACTLY args.steps steps ----
    step_times = []
    t0 = time.perf_counter()
    for i in range(args.steps):
        ts = time.perf_counter()
        step(args.warmup + i)
        sync()
        step_times.append(time.perf_counter() - ts)
    sync()
    barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if dist is not None:
        e = torch.tensor([elapsed], dtype=torch.float64, device=device if use_gpu else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    total_lines = args.batch * args.steps * world_size
    lines_per_sec = total_lines / elapsed
    ms_per_step = elapsed / args.steps * 1000.0
    step_sorted = sorted(step_times)
    p50_step = step_sorted[len(step_sorted) // 2]
    p99_step = step_sorted[min(len(step_sorted) - 1, int(len(step_sorted) * 0.99))]
    p50_line_us = p50_step / args.batch * 1e6

    if rank == 0:
        result = {
            "metric": "lines_per_sec",
            "value": round(lines_per_sec, 1),
            "unit": "lines/s",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "matcher_parser+new_value+bert-tiny-detector",
                "global_batch": args.batch * world_size,
                "seq_len": 64,
                "max_line_len": args.max_len,
                "parallelism": f"dp{world_size}",
                "p50_detect_latency_us_per_line": round(p50_line_us, 3),
                "p50_batch_ms": round(p50_step * 1000.0, 3),
                "p99_batch_ms": round(p99_step * 1000.0, 3),
                "transformer": not args.no_transformer,
                "hip_graph": use_graph[0],
                "device": str(device),
            },
        }
        print(json.dumps(result))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
