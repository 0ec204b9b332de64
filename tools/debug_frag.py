"""Isolate frag-chain pieces: fragify round-trip, XF GEMM vs std,
gu frag in/out, attention fragout, qkv-rope XF."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from ollamamq_amd.ops import hip
hip.require()

def frag_ref(x):   # [M,K] -> frag flat [32*K] (fp gold)
    M, K = x.shape
    out = torch.zeros(32 * K, dtype=x.dtype, device=x.device)
    xr = x.view(M, K // 64, 4, 2, 8)   # m, b, j, h, e
    for m in range(M):
        for b in range(K // 64):
            for j in range(4):
                for h in range(2):
                    u = ((b * 4 + j) * 64) + h * 32 + m
                    out[u * 8:(u + 1) * 8] = xr[m, b, j, h]
    return out

g = torch.Generator().manual_seed(3)
M, K, N = 5, 512, 1024
x = (torch.randn(M, K, generator=g) * 0.5).bfloat16().cuda()

# 1. fragify matches reference mapping
xf, sq = hip.fragify_sumsq(x)
ref = frag_ref(x)
mask = torch.zeros(32 * K, dtype=torch.bool)
for m in range(M):
    for u0 in range(K // 8):
        b, j, h = u0 // 8, (u0 // 2) % 4, u0 % 2
        mask[(((b * 4 + j) * 64) + h * 32 + m) * 8:][:8] = True
ok1 = torch.equal(xf.cpu()[mask], ref.cpu()[mask])
print("fragify:", ok1)
print("sumsq:", torch.allclose(sq.cpu(), (x.float().cpu() ** 2).sum(-1),
                               rtol=1e-2))

# 2. XF GEMM == std GEMM
w = (torch.randn(N, K, generator=g) * 0.1).bfloat16().cuda()
pk = hip.pack_weight(w)
y_std = hip.linear_packed(x, pk, None, N)
y_xf = hip.linear_packed(xf, pk, None, N, K=K, xlds=2)
print("xf-gemm:", torch.allclose(y_xf[:M].float(), y_std.float(),
                                 atol=5e-2, rtol=5e-2))

# 3. yfrag output round-trip: emit frag, defragify by reading ref-mapped
y_fr = hip.linear_packed(xf, pk, None, N, K=K, xlds=2, yfrag=1)
# y_fr holds frag layout over N; compare element (m, n)
got = torch.empty(M, N)
for m in range(M):
    for n in range(N):
        b, j, h, e = n // 64, (n // 16) % 4, (n // 8) % 2, n % 8
        u = ((b * 4 + j) * 64) + h * 32 + m
        got[m, n] = y_fr.view(-1)[u * 8 + e].float()
print("yfrag:", torch.allclose(got, y_std.float().cpu(), atol=5e-2,
                               rtol=5e-2))

# 4. gu frag in+out
F = 512
wg = (torch.randn(2 * F, K, generator=g) * 0.1).bfloat16().cuda()
gpk = hip.pack_weight_gu(wg)
act_std = hip.linear_gu(x, gpk, 2 * F)
act_fr = hip.linear_gu(xf, gpk, 2 * F, K=K, yfrag=1)
got = torch.empty(M, F)
for m in range(M):
    for n in range(F):
        b, j, h, e = n // 64, (n // 16) % 4, (n // 8) % 2, n % 8
        u = ((b * 4 + j) * 64) + h * 32 + m
        got[m, n] = act_fr.view(-1)[u * 8 + e].float()
print("gu:", torch.allclose(got, act_std.float().cpu(), atol=5e-2,
                            rtol=5e-2))

# 5. attention fragout vs std
from ollamamq_amd.models import PRESETS, LlamaModel
from ollamamq_amd.engine import PagedKVCache
from ollamamq_amd.ops.interface import AttnMeta
cfg = PRESETS["tiny"]
kv = PagedKVCache.for_model(cfg, n_pages=64, max_slots=4, max_ctx=128,
                            device="cuda", dtype=torch.bfloat16)
B = 3
slots = [kv.alloc_slot() for _ in range(B)]
L = 9
for s_ in slots:
    kv.ensure(s_, L)
kv.k_pool.normal_(0, 0.3)
kv.v_pool.normal_(0, 0.3)
q = (torch.randn(B, cfg.n_heads, 128, generator=g) * 0.3).bfloat16().cuda()
meta = AttnMeta(mode="decode",
                slot_ids=torch.tensor(slots, dtype=torch.int32,
                                      device="cuda"),
                seq_lens=torch.tensor([L] * B, dtype=torch.int32,
                                      device="cuda"),
                cu_q=torch.arange(B + 1, dtype=torch.int32, device="cuda"),
                logits_idx=None, max_q=1, max_kv=L, window=0)
a_std = hip.attention_decode(q, kv, 0, meta)
a_fr = hip.attention_decode(q, kv, 0, meta, fragout=True)
Kh = cfg.n_heads * 128
got = torch.empty(B, Kh)
for m in range(B):
    for n in range(Kh):
        b, j, h, e = n // 64, (n // 16) % 4, (n // 8) % 2, n % 8
        u = ((b * 4 + j) * 64) + h * 32 + m
        got[m, n] = a_fr.view(-1)[u * 8 + e].float()
print("attn-frag:", torch.allclose(got, a_std.view(B, Kh).float().cpu(),
                                   atol=5e-2, rtol=5e-2))
