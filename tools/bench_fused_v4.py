#!/usr/bin/env python3
"""A/B the production (2 blocks/CU) vs v4 occupancy (3 blocks/CU) fused
BERT kernels."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from detectmateservice_amd.models.bert_tiny import BertTinyDetectorModel
from detectmateservice_amd.ops import _dmx_C


def bench(fn, args, iters=30, warm=5):
    for _ in range(warm):
        fn(*args)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn(*args)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    model = BertTinyDetectorModel(device="cuda")
    wb, fb = model._fused_blobs()
    for B in (4096, 16384, 65536, 131072):
        lines = torch.randint(32, 127, (B, 256), dtype=torch.uint8, device="cuda")
        start = torch.zeros(B, dtype=torch.int32, device="cuda")
        end = torch.full((B,), 180, dtype=torch.int32, device="cuda")
        a = (lines, start, end, wb, fb, 2, 1e-5)
        t3 = bench(_dmx_C.bert_fused_bf16, a)
        t4 = bench(_dmx_C.bert_fused_bf16_v4, a)
        print(f"B={B:7d}  v3={t3:8.3f} ms ({B/t3*1000/1e6:6.2f}M l/s)  "
              f"v4={t4:8.3f} ms ({B/t4*1000/1e6:6.2f}M l/s)  "
              f"speedup={t3/t4:5.2f}x", flush=True)


if __name__ == "__main__":
    main()
