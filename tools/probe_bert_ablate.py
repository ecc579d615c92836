"""Perf-ablation of the fused kernel's GEMM operand loads (ABL bits in
the probe mask high byte): which in-phase waits dominate?
  +256 = B (L2 weight) loads replaced by an opaque register
  +512 = A (LDS activation) loads replaced by an opaque register
  +768 = both (pure MFMA + epilogue skeleton)
Numerically wrong by design — timing diagnostics only."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from detectmateservice_amd import ops
from detectmateservice_amd.models.bert_tiny import BertTinyDetectorModel, BertTinyConfig
from detectmateservice_amd.ops import _dmx_C
from detectmateservice_amd.utils.synthetic import AuditLogGenerator

gen = AuditLogGenerator(seed=1)
B = 65536
raw = [gen.line()[0].encode() for _ in range(B)]
lines, lens = ops.pack_lines(raw, 256, device="cuda")
start = torch.zeros(B, dtype=torch.int32, device="cuda")
m = BertTinyDetectorModel(BertTinyConfig(), device="cuda")
wb, fb = m._fused_blobs()

def bench(mask, iters=10):
    for _ in range(3):
        _dmx_C.bert_fused_probe(lines, start, lens.int(), wb, fb, 2, 1e-5, mask)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _dmx_C.bert_fused_probe(lines, start, lens.int(), wb, fb, 2, 1e-5, mask)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000

names = {31: "ALL (production path)", 31+256: "ALL, B-const (no L2 wloads)",
         31+512: "ALL, A-const (no LDS reads)", 31+768: "ALL, both const",
         16: "FFN only", 16+256: "FFN, B-const", 16+512: "FFN, A-const",
         16+768: "FFN, both const"}
for mask in (31, 31+256, 31+512, 31+768, 16, 16+256, 16+512, 16+768):
    t = bench(mask)
    print(f"mask={mask:4d} ({names[mask]:30s}): {t:7.3f} ms")
