"""Isolate the service >=32768 cliff: time the exact alloc/copy/free
cycle the packed loop drives — a [B, 256] u8 pinned tensor allocated per
batch, H2D'd async, held one extra iteration (pipelined depth 1), then
freed — for B around the cliff. If 32768 cycles are ~10x slower than
24576, the torch host caching allocator is re-pinning each batch
(hipHostMalloc device-syncs) and explicit double-buffered staging is
the fix."""
import sys
import time

import torch


def cycle(B, pin, iters=40):
    dev = "cuda"
    held = []
    # warm
    for _ in range(4):
        t = torch.empty((B, 256), dtype=torch.uint8, pin_memory=pin)
        d = t.to(dev, non_blocking=True)
        held.append((t, d))
        if len(held) > 2:
            held.pop(0)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        t = torch.empty((B, 256), dtype=torch.uint8, pin_memory=pin)
        d = t.to(dev, non_blocking=True)
        held.append((t, d))
        if len(held) > 2:
            held.pop(0)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    for B in (16384, 24576, 28672, 32768, 36864, 49152, 65536):
        ms_pin = cycle(B, True)
        ms_page = cycle(B, False)
        print(f"B={B:6d}  pinned {ms_pin:7.3f} ms/cycle   "
              f"pageable {ms_page:7.3f} ms/cycle", flush=True)


if __name__ == "__main__":
    main()
