import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from detectmateservice_amd import ops
from detectmateservice_amd.models.bert_tiny import BertTinyDetectorModel, BertTinyConfig
from detectmateservice_amd.ops import _dmx_C
from detectmateservice_amd.utils.synthetic import AuditLogGenerator
gen = AuditLogGenerator(seed=1)
B = 32768
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
names = {0:"none", 1:"qkv", 3:"qkv+attn", 7:"+proj", 15:"+ln", 31:"ALL", 16:"ffn only", 8:"ln only"}
base = None
for mask in (0, 1, 3, 7, 15, 31, 16, 8):
    t = bench(mask)
    print(f"mask={mask:2d} ({names[mask]:9s}): {t:7.3f} ms")
# sanity: probe(31) == production kernel scores
s1 = _dmx_C.bert_fused_probe(lines, start, lens.int(), wb, fb, 2, 1e-5, 31)
s2 = _dmx_C.bert_fused_bf16(lines, start, lens.int(), wb, fb, 2, 1e-5)
print("probe==prod:", bool(torch.allclose(s1, s2)))
