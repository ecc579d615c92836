"""Regime probe: the SAME BertTiny forward as a LAYERED batched model on
plain torch/rocBLAS (matmul at M=B*S) vs the per-line megakernel.
At B=65536 each GEMM is [4.2M x 128] @ [128 x N] — dense-MFMA regime.
Numbers decide whether the flagship backbone should be layered."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

B, S, H, HEADS, FFN, LAYERS = 65536, 64, 128, 2, 512, 2
dev = "cuda"
dt = torch.bfloat16
g = torch.Generator().manual_seed(0)
def rnd(*sh): return (torch.randn(*sh, generator=g) * 0.02).to(dt).to(dev)

tok_emb = rnd(259, H); pos_emb = rnd(S, H)
layers = []
for _ in range(LAYERS):
    layers.append(dict(
        wqkv=rnd(H, 3 * H), bqkv=rnd(3 * H).float(),
        wo=rnd(H, H), bo=rnd(H).float(),
        w1=rnd(H, FFN), b1=rnd(FFN).float(),
        w2=rnd(FFN, H), b2=rnd(H).float(),
        g1=rnd(H).float(), be1=rnd(H).float(),
        g2=rnd(H).float(), be2=rnd(H).float(),
    ))
w_score = rnd(H).float(); b_score = 0.1

tokens = torch.randint(0, 259, (B, S), device=dev)

def fwd(tokens):
    x = tok_emb[tokens] + pos_emb  # [B, S, H]
    for L in layers:
        qkv = (x.reshape(-1, H) @ L["wqkv"]).reshape(B, S, 3 * H) + L["bqkv"].to(dt)
        q, k, v = qkv.split(H, dim=2)
        q = q.view(B, S, HEADS, H // HEADS).transpose(1, 2)
        k = k.view(B, S, HEADS, H // HEADS).transpose(1, 2)
        v = v.view(B, S, HEADS, H // HEADS).transpose(1, 2)
        att = torch.softmax((q @ k.transpose(-1, -2)) * (1.0 / 8.0), dim=-1)
        o = (att @ v).transpose(1, 2).reshape(B, S, H)
        x = x + (o.reshape(-1, H) @ L["wo"]).reshape(B, S, H) + L["bo"].to(dt)
        x = torch.nn.functional.layer_norm(x.float(), (H,), L["g1"], L["be1"]).to(dt)
        hdn = torch.nn.functional.gelu(
            (x.reshape(-1, H) @ L["w1"]).reshape(B, S, FFN) + L["b1"].to(dt),
            approximate="tanh")
        x = x + (hdn.reshape(-1, FFN) @ L["w2"]).reshape(B, S, H) + L["b2"].to(dt)
        x = torch.nn.functional.layer_norm(x.float(), (H,), L["g2"], L["be2"]).to(dt)
    return x.mean(dim=1).float() @ w_score + b_score

for _ in range(3):
    fwd(tokens)
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 10
for _ in range(iters):
    fwd(tokens)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / iters * 1000
print(f"torch layered forward: {ms:.3f} ms for B={B} ({B/ms*1000:,.0f} lines/s)")

# also try scaled_dot_product_attention (may use fused flash path)
def fwd_sdpa(tokens):
    x = tok_emb[tokens] + pos_emb
    for L in layers:
        qkv = (x.reshape(-1, H) @ L["wqkv"]).reshape(B, S, 3 * H) + L["bqkv"].to(dt)
        q, k, v = qkv.split(H, dim=2)
        q = q.view(B, S, HEADS, H // HEADS).transpose(1, 2)
        k = k.view(B, S, HEADS, H // HEADS).transpose(1, 2)
        v = v.view(B, S, HEADS, H // HEADS).transpose(1, 2)
        o = torch.nn.functional.scaled_dot_product_attention(q, k, v)
        o = o.transpose(1, 2).reshape(B, S, H)
        x = x + (o.reshape(-1, H) @ L["wo"]).reshape(B, S, H) + L["bo"].to(dt)
        x = torch.nn.functional.layer_norm(x.float(), (H,), L["g1"], L["be1"]).to(dt)
        hdn = torch.nn.functional.gelu(
            (x.reshape(-1, H) @ L["w1"]).reshape(B, S, FFN) + L["b1"].to(dt),
            approximate="tanh")
        x = x + (hdn.reshape(-1, FFN) @ L["w2"]).reshape(B, S, H) + L["b2"].to(dt)
        x = torch.nn.functional.layer_norm(x.float(), (H,), L["g2"], L["be2"]).to(dt)
    return x.mean(dim=1).float() @ w_score + b_score

for _ in range(3):
    fwd_sdpa(tokens)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(iters):
    fwd_sdpa(tokens)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / iters * 1000
print(f"torch layered (sdpa):  {ms:.3f} ms for B={B} ({B/ms*1000:,.0f} lines/s)")
