#!/usr/bin/env python3
"""Flagship benchmark: lines/sec through reader→parser→detector.

Measures the BASELINE.json headline metric — synthetic audit-log lines
flowing through the full ingest + GPU pipeline — on 1..N MI355X GPUs,
one rank per GPU, data-parallel detectors with an RCCL all-gather of
per-rank anomaly summaries each step (BASELINE config 4).

The timed region covers the WHOLE reader→parser→detector path
(VERDICT round-1 item 4): serialized LogSchema protobuf frames travel
through a /dev/shm ring (the service's shm:// transport), are
proto-decoded IN PLACE into pinned [B, max_len] tensors by C++ reader
threads (GIL released — ops/csrc/shm_ring.cpp), copied to HBM, and run
through the fused template-match → NewValue hash-probe → BERT-tiny bf16
MFMA pipeline. Decode of step N+1 overlaps step N's GPU work exactly
like the production packed engine loop. Nothing is cached inside the
timed region: every step re-decodes protobuf frames and runs every GPU
stage. ``--ingest off`` falls back to the round-1 device-resident-pool
measurement (GPU pipeline only, for kernel-regression tracking).

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for
N>1 the driver launches via torch.distributed.run with one rank per GPU.
W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides; elapsed = MAX over ranks;
rank 0 prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import queue
import threading
import time
import uuid

import torch

from detectmateservice_amd import ops
from detectmateservice_amd.pipeline import GpuPipeline, PipelineConfig
from detectmateservice_amd.schemas import LogSchema
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
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-capture the steady-state step (ingest off only)")
    p.add_argument("--ingest", choices=["shm", "off"], default="shm",
                   help="shm: protobuf frames through the shm ring with C++ "
                        "in-place decode (full reader path, default); "
                        "off: device-resident line pool (GPU pipeline only)")
    p.add_argument("--shards", type=int, default=0,
                   help="ingest shards (ring + feeder + reader threads) per "
                        "rank; 0 = auto (4 single-rank, 2 when several ranks "
                        "share the node's cores — measured equal-throughput "
                        "single-rank, half the thread pressure at dp8)")
    p.add_argument("--chunk", type=int, default=16384,
                   help="frames per ring read (constant size so pinned "
                        "buffers come from the caching allocator)")
    p.add_argument("--line-buffer", action="store_true",
                   help="retain every ingested batch in the capacity-managed "
                        "HBM line buffer (BASELINE config 5) during the run")
    p.add_argument("--device", default=None)
    return p.parse_args()


class ShardFeeder(threading.Thread):
    """Writes pre-serialized LogSchema frames into a shm ring (C++
    write_frames releases the GIL; drop-don't-block partial writes are
    retried — the reader applies backpressure through the ring)."""

    def __init__(self, ring, frame_pool, total_frames):
        super().__init__(daemon=True)
        self.ring = ring
        self.pool = frame_pool
        self.total = total_frames

    def run(self):
        sent = 0
        i = 0
        while sent < self.total:
            batch = self.pool[i % len(self.pool)]
            i += 1
            need = min(len(batch), self.total - sent)
            frames = batch[:need]
            off = 0
            while off < len(frames):
                n = self.ring.write_frames(frames[off:] if off else frames)
                if n <= 0:
                    time.sleep(0.0002)
                    continue
                off += n
            sent += len(frames)


class ShardReader(threading.Thread):
    """Drains a ring into decoded pinned tensors, one step's worth at a
    time (GIL-released C++ scan + protobuf decode, shm_ring.cpp)."""

    def __init__(self, ring, lines_per_step, n_steps, chunk, max_len, out_q,
                 pin, device=None, copy_stream=None):
        super().__init__(daemon=True)
        self.ring = ring
        self.lines_per_step = lines_per_step
        self.n_steps = n_steps
        self.chunk = chunk
        self.max_len = max_len
        self.out_q = out_q
        self.pin = pin
        self.device = device
        self.copy_stream = copy_stream

    def run(self):
        # per-CHUNK queueing: the GPU starts on a step's first decoded
        # chunk while this thread is still decoding the rest of it.
        # On GPU the H2D upload happens HERE (eagerly, on the shared
        # copy stream) so every chunk of step N+1 lands on-device during
        # step N's compute — the step boundary no longer exposes the
        # last chunk's decode+copy tail.
        try:
            if self.copy_stream is not None:
                torch.cuda.set_device(self.device)
            for _ in range(self.n_steps):
                got = 0
                while got < self.lines_per_step:
                    want = min(self.chunk, self.lines_per_step - got)
                    lines, lens, blob, off, nb = self.ring.read_batch_packed(
                        want, 100, self.max_len, self.pin)
                    b = int(lines.shape[0])
                    if b == 0:
                        continue
                    if self.copy_stream is not None:
                        with torch.cuda.stream(self.copy_stream):
                            dl = lines.to(self.device, non_blocking=True)
                            dn = lens.to(self.device, non_blocking=True)
                            ev = torch.cuda.Event()
                            ev.record(self.copy_stream)
                        # keep the host (staging) tensors referenced
                        # until the consumer observed the copy event
                        self.out_q.put((dl, dn, ev, lines, lens))
                    else:
                        self.out_q.put((lines, lens))
                    got += b
                self.out_q.put("step_end")
        except Exception as exc:  # noqa: BLE001 - surface in the main loop
            self.out_q.put(exc)


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

    ingest = args.ingest
    if ingest == "shm" and not ops.have_extension():
        print("# extension missing: falling back to --ingest off", flush=True)
        ingest = "off"

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

    line_buf = None
    if args.line_buffer:
        from detectmateservice_amd.line_buffer import GpuLineBuffer

        # budget_fraction of FREE HBM after model/pools (288 GB part:
        # >500M resident lines at max_len 256)
        line_buf = GpuLineBuffer(max_len=args.max_len, device=device,
                                 budget_fraction=0.5)

    gen = AuditLogGenerator(seed=1000 + rank, anomaly_rate=0.01)

    use_graph = [False]
    n_total_steps = args.warmup + args.steps

    if ingest == "shm":
        from detectmateservice_amd.ops import _dmx_C

        n_shards = args.shards or (4 if world_size == 1 else 2)
        shards = max(1, min(n_shards, args.batch))  # no zero-line shards
        per_shard = args.batch // shards
        rem = args.batch - per_shard * shards
        shard_lines = [per_shard + (1 if i < rem else 0) for i in range(shards)]
        rings, feeders, readers, queues = [], [], [], []
        ring_bytes = 64 << 20
        run_id = uuid.uuid4().hex[:8]
        # H2D copies ride a dedicated stream, driven by the READER
        # threads right after decode: chunk N+1's upload overlaps chunk
        # N's kernels and the step boundary exposes no copy tail
        copy_stream = torch.cuda.Stream(device) if use_gpu else None
        for i, nlines in enumerate(shard_lines):
            path = f"/dev/shm/dmx-bench-{run_id}-r{rank}s{i}"
            ring = _dmx_C.ShmRing(path, ring_bytes, True)
            # chunks sit in a depth-8 queue + upload/compute in flight:
            # the ring's rotating staging must outlive them all
            ring.set_staging_depth(12)
            # per-shard frame pool: distinct pre-serialized protobuf
            # batches (generation + serialization are reader work the
            # reference does outside the service too — the TIMED work is
            # transport + decode + parse + detect)
            pool = []
            for _ in range(args.pool):
                pool.append([
                    LogSchema(logID=f"l{j}", log=gen.line()[0]).serialize()
                    for j in range(nlines)
                ])
            q: "queue.Queue" = queue.Queue(maxsize=8)
            feeders.append(ShardFeeder(ring, pool, nlines * n_total_steps))
            readers.append(ShardReader(ring, nlines, n_total_steps,
                                        args.chunk, args.max_len, q, use_gpu,
                                        device if use_gpu else None,
                                        copy_stream))
            rings.append((ring, path))
            queues.append(q)
        for f in feeders:
            f.start()
        for r in readers:
            r.start()

        def step(i: int) -> None:
            outs = []
            for q in queues:
                while True:
                    item = q.get(timeout=120)
                    if isinstance(item, Exception):
                        raise item
                    if item == "step_end":
                        break
                    if len(item) == 5:  # GPU: reader already uploaded
                        dl, dn, ev, _hl, _hn = item
                        torch.cuda.current_stream(device).wait_event(ev)
                    else:  # CPU fallback: host tensors
                        lines, lens = item
                        dl, dn = lines.to(device), lens.to(device)
                    if line_buf is not None:
                        line_buf.append(dl, dn)
                    outs.append(pipe.process_packed(dl, dn))
            if dist is not None:
                # DP aggregation over RCCL/xGMI — the same helper the
                # Service dp path uses (parallel/dist.py)
                from detectmateservice_amd.parallel import dist as dmx_dist

                summary = torch.stack([
                    sum(o["anomaly"].sum() for o in outs).float(),
                    sum(o["scores"].sum() for o in outs),
                ]).to(device)
                dmx_dist.all_gather_summaries(summary)

        def cleanup():
            for _, path in rings:
                try:
                    os.unlink(path)
                except OSError:
                    pass
    else:
        # device-resident pool (round-1 measurement: GPU pipeline only)
        pool_dev = []
        for _ in range(args.pool):
            raw = [gen.line()[0].encode() for _ in range(args.batch)]
            lines, lens = ops.pack_lines(raw, args.max_len, device="cpu")
            pool_dev.append((lines.to(device), lens.to(device)))

        def step(i: int) -> None:
            lines, lens = pool_dev[i % len(pool_dev)]
            if line_buf is not None:
                line_buf.append(lines, lens)
            if use_graph[0]:
                out = pipe.process_packed_graph(lines, lens)
            else:
                out = pipe.process_packed(lines, lens)
            if dist is not None:
                from detectmateservice_amd.parallel import dist as dmx_dist

                summary = torch.stack(
                    [out["anomaly"].sum().float(), out["scores"].sum()]
                ).to(device)
                dmx_dist.all_gather_summaries(summary)

        def cleanup():
            pass

    def sync():
        if use_gpu:
            torch.cuda.synchronize()

    def barrier():
        if dist is not None:
            dist.barrier()

    # ---- warmup (also trains the NewValue hash sets on the first batch) ----
    for i in range(args.warmup):
        step(i)
    sync()
    if dist is not None and pipe.hashsets is not None and use_gpu:
        # BASELINE config-4 semantics end-to-end: merge the data-parallel
        # NewValue hash sets once after training — a REAL RCCL all-gather
        # + batched insert over xGMI in the driver's multi-GPU runs
        # (outside the timed region; parallel/dist.py::all_reduce_hashsets)
        from detectmateservice_amd.parallel import dist as dmx_dist

        dmx_dist.all_reduce_hashsets(pipe.hashsets.tables)
        sync()
    if use_gpu and args.graph and ingest == "off":
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

    cleanup()

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
                "ingest": ("shm_ring+proto_decode" if ingest == "shm" else
                           "device_pool(gpu_only)"),
                "p50_detect_latency_us_per_line": round(p50_line_us, 3),
                "p50_batch_ms": round(p50_step * 1000.0, 3),
                "p99_batch_ms": round(p99_step * 1000.0, 3),
                "transformer": not args.no_transformer,
                "hip_graph": use_graph[0],
                "line_buffer": (line_buf.watermark() if line_buf is not None
                                else None),
                "device": str(device),
            },
        }
        print(json.dumps(result))

    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
