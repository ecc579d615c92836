"""Single-process service simulator for the engine-batch >=32768 cliff.

Reproduces the packed engine cadence exactly — C++ ShmFeeder → ring →
read_batch_packed(pin) → H2D → GpuPipeline.process_packed, pipelined
depth 1 with the anomaly readback as the collect sync — with flags that
knock out one layer at a time. Run on GPU with a batch above and below
the cliff per mode; the mode that stays fast names the culprit layer.

Measured layers already individually FLAT at 32768 (no cliff):
kernels in isolation (probe_batch_cliff), pinned alloc+H2D cycles
(probe_pinned_cycle), saturated-ring decode on CPU (ring_cliff2).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from detectmateservice_amd.ops import _dmx_C
from detectmateservice_amd.pipeline import GpuPipeline, PipelineConfig
from detectmateservice_amd.schemas import LogSchema
from detectmateservice_amd.utils.synthetic import (
    AUDIT_LOG_FORMAT,
    AUDIT_TEMPLATES,
    AuditLogGenerator,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=32768)
    ap.add_argument("--lines", type=int, default=3_000_000)
    ap.add_argument("--pin", type=int, default=1)
    ap.add_argument("--pipeline", type=int, default=1,
                    help="1 = depth-1 pipelining like the engine")
    ap.add_argument("--h2d", type=int, default=1,
                    help="0 = score preloaded device tensors instead")
    ap.add_argument("--kernels", type=int, default=1,
                    help="0 = decode+H2D only, skip the pipeline")
    args = ap.parse_args()

    dev = "cuda"
    gen = AuditLogGenerator(seed=1, anomaly_rate=0.0)
    frames = [LogSchema(logID=f"l{i}", log=gen.line()[0]).serialize()
              for i in range(8192)]
    path = "/dev/shm/dmx-sim-svc"
    RING = 32 << 20
    try:
        os.unlink(path)
    except OSError:
        pass
    ring = _dmx_C.ShmRing(path, RING, True)
    feeder = _dmx_C.ShmFeeder(path, frames, RING)

    pipe = static_l = static_n = None
    if args.kernels:
        cfg = PipelineConfig(
            templates=list(AUDIT_TEMPLATES), log_format=AUDIT_LOG_FORMAT,
            watches=[{"kind": "variable", "pos": 5, "event": 1}],
            train_lines=0, use_transformer=True, score_threshold=1.0e9,
            max_len=256)
        pipe = GpuPipeline(cfg, device=dev)
        static_l = torch.randint(32, 127, (args.batch, 256),
                                 dtype=torch.uint8, device=dev)
        static_n = torch.full((args.batch,), 150, dtype=torch.int32,
                              device=dev)

    def submit(lines_cpu, lens_cpu):
        if args.h2d:
            dl = lines_cpu.to(dev, non_blocking=True)
            dn = lens_cpu.to(dev, non_blocking=True)
        elif args.kernels:
            B = lines_cpu.shape[0]
            dl, dn = static_l[:B], static_n[:B]
        else:
            return (None, lines_cpu)  # pure CPU loop (no CUDA at all)
        if not args.kernels:
            return (dl, lines_cpu)
        return (pipe.process_packed(dl, dn), lines_cpu)

    def collect(token):
        out, _held = token
        if args.kernels:
            bool(out["anomaly"].any())  # the engine's alert readback sync
        elif args.h2d:
            torch.cuda.synchronize()

    feeder.start(args.lines)
    time.sleep(0.8)  # saturate the ring like the live service
    got = reads = 0
    t_read = t_sub = t_col = 0.0
    prev = None
    t0 = time.perf_counter()
    while got < args.lines:
        r0 = time.perf_counter()
        lines_cpu, lens_cpu, blob, off, nbytes = ring.read_batch_packed(
            args.batch, 200, 256, bool(args.pin))
        r1 = time.perf_counter()
        t_read += r1 - r0
        if lines_cpu.shape[0] == 0:
            continue
        got += lines_cpu.shape[0]
        reads += 1
        if args.pipeline:
            if prev is not None:
                c0 = time.perf_counter()
                collect(prev)
                t_col += time.perf_counter() - c0
            s0 = time.perf_counter()
            prev = submit(lines_cpu, lens_cpu)
            t_sub += time.perf_counter() - s0
        else:
            s0 = time.perf_counter()
            tok = submit(lines_cpu, lens_cpu)
            collect(tok)
            t_sub += time.perf_counter() - s0
    if prev is not None:
        collect(prev)
    dt = time.perf_counter() - t0
    feeder.join(2000)
    print(f"batch={args.batch} pin={args.pin} pipe={args.pipeline} "
          f"h2d={args.h2d} kern={args.kernels}: "
          f"{args.lines/dt/1e6:6.2f} M lines/s  "
          f"read {t_read/reads*1e3:5.2f} sub {t_sub/reads*1e3:5.2f} "
          f"col {t_col/max(reads,1)*1e3:5.2f} ms/b ({reads} reads)",
          flush=True)


if __name__ == "__main__":
    main()
