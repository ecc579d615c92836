#!/usr/bin/env python3
"""Per-wave phase timing of the fused kernel: work vs barrier-wait split
(stamps from s_memrealtime, 100 MHz constant clock -> 10 ns units)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from detectmateservice_amd import ops  # noqa: E402
from detectmateservice_amd.models.bert_tiny import (  # noqa: E402
    BertTinyConfig,
    BertTinyDetectorModel,
)
from detectmateservice_amd.ops import _dmx_C  # noqa: E402
from detectmateservice_amd.utils.synthetic import AuditLogGenerator  # noqa: E402

B = 32768
gen = AuditLogGenerator(seed=1)
raw = [gen.line()[0].encode() for _ in range(B)]
lines, lens = ops.pack_lines(raw, 256, device="cuda")
start = torch.zeros(B, dtype=torch.int32, device="cuda")
m = BertTinyDetectorModel(BertTinyConfig(), device="cuda")
wb, fb = m._fused_blobs()

for _ in range(3):
    _dmx_C.bert_fused_timed(lines, start, lens.int(), wb, fb, 2, 1e-5)
torch.cuda.synchronize()
scores, stamps = _dmx_C.bert_fused_timed(lines, start, lens.int(), wb, fb, 2, 1e-5)
torch.cuda.synchronize()
st = stamps.cpu().double() * 10.0  # ns (100 MHz ticks)

# slots: 0 start, (1,2) embed, per layer l: (3+8l..): qkv(3,4) attn(5,6)
# projLN(7,8) ffnLN(9,10); layer1: 11..18; 19 pool/final... actual count:
# 2 + 1 + 2*8 + 1 = 20 used of 22
names = ["embed"]
for l in range(2):
    names += [f"L{l}.qkv", f"L{l}.attn", f"L{l}.projLN", f"L{l}.ffnLN"]
names += ["pool"]

# per (block, wave): work_k = t[2k+1]-t[2k?]... reconstruct indices:
# t0 start; pairs: (1,2),(3,4),...(17,18); final 19
work_tot = {}
wait_tot = {}
prev_cross = st[..., 0]
for k, name in enumerate(names[:-1]):
    w = st[..., 1 + 2 * k] - prev_cross          # work for this phase
    wait = st[..., 2 + 2 * k] - st[..., 1 + 2 * k]  # barrier wait
    prev_cross = st[..., 2 + 2 * k]
    work_tot[name] = w
    wait_tot[name] = wait
work_tot["pool"] = st[..., 19] - prev_cross

blk_span = st[..., 19].amax(dim=1) - st[..., 0].amin(dim=1)
print(f"block span: mean {blk_span.mean()/1e3:.2f} us  p95 {blk_span.quantile(0.95)/1e3:.2f} us")
tot_work = sum(v.mean() for v in work_tot.values())
tot_wait = sum(v.mean() for v in wait_tot.values())
print(f"mean per-wave: WORK {tot_work/1e3:.2f} us  BARRIER-WAIT {tot_wait/1e3:.2f} us")
print(f"{'phase':10s} {'work us (mean/p95)':>22s} {'wait us (mean/p95)':>22s}")
for name in work_tot:
    w = work_tot[name]
    a = wait_tot.get(name)
    ws = f"{w.mean()/1e3:6.2f} / {w.quantile(0.95)/1e3:6.2f}"
    as_ = f"{a.mean()/1e3:6.2f} / {a.quantile(0.95)/1e3:6.2f}" if a is not None else "-"
    print(f"{name:10s} {ws:>22s} {as_:>22s}")
