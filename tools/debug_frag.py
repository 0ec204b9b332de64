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

# ---- MT2 (rows 33..64) variants ----
print("--- MT2 ---")
M2 = 40
x2 = (torch.randn(M2, K, generator=g) * 0.5).bfloat16().cuda()
xf2, sq2 = hip.fragify_sumsq(x2)
assert xf2.numel() == 64 * K
print("fragify64 sumsq:", torch.allclose(
    sq2.cpu(), (x2.float().cpu() ** 2).sum(-1), rtol=1e-2))

def defrag2(flat, M, Kd):
    got = torch.empty(M, Kd)
    fv = flat.reshape(-1).float().cpu()
    for m in range(M):
        half = (m // 32) * 32 * Kd
        mr = m % 32
        for n in range(Kd):
            b, j, h, e = n // 64, (n // 16) % 4, (n // 8) % 2, n % 8
            u = ((b * 4 + j) * 64) + h * 32 + mr
            got[m, n] = fv[half + u * 8 + e]
    return got

print("fragify64 layout:", torch.allclose(
    defrag2(xf2, M2, K), x2.float().cpu(), atol=1e-3))

y2_std = hip.linear_packed(x2, pk, None, N)
y2_xf = hip.linear_packed(xf2, pk, None, N, K=K, xlds=2, M_frag=M2)
print("mt2 xf-gemm:", torch.allclose(y2_xf[:M2].float(), y2_std.float(),
                                     atol=5e-2, rtol=5e-2))
y2_fr = hip.linear_packed(xf2, pk, None, N, K=K, xlds=2, yfrag=1,
                          M_frag=M2)
print("mt2 yfrag:", torch.allclose(defrag2(y2_fr, M2, N),
                                   y2_std.float().cpu(), atol=5e-2,
                                   rtol=5e-2))
act2_std = hip.linear_gu(x2, gpk, 2 * F)
act2_fr = hip.linear_gu(xf2, gpk, 2 * F, K=K, yfrag=1, M_frag=M2)
print("mt2 gu:", torch.allclose(defrag2(act2_fr, M2, F),
                                act2_std.float().cpu(), atol=5e-2,
                                rtol=5e-2))
# ks>1 path (down-like) + res + sq
N3, K3 = 512, 14336
w3 = (torch.randn(N3, K3, generator=g) * 0.05).bfloat16().cuda()
pk3 = hip.pack_weight(w3)
x3 = (torch.randn(M2, K3, generator=g) * 0.3).bfloat16().cuda()
xf3, _ = hip.fragify_sumsq(x3)
res3 = (torch.randn(M2, N3, generator=g) * 0.3).bfloat16().cuda()
rf3, _ = hip.fragify_sumsq(res3)
sqo = torch.zeros(64 * (N3 // 32), dtype=torch.float32, device="cuda")
hip.linear_packed(xf3, pk3, None, N3, K=K3, xlds=2, yfrag=1,
                  M_frag=M2, res=rf3, sq_out=sqo, y=rf3)
ref3 = (torch.nn.functional.linear(x3.float(), w3.float())
        + res3.float()).cpu()
print("mt2 ks res+y:", torch.allclose(defrag2(rf3, M2, N3), ref3,
                                      atol=8e-2, rtol=8e-2))
sq_ref = (defrag2(rf3, M2, N3) ** 2).view(M2, N3 // 32, 32).sum(-1)
sq_got = sqo.view(64, N3 // 32)[:M2].cpu()
print("mt2 sq:", torch.allclose(sq_got, sq_ref, rtol=5e-2, atol=5e-2))


# ---- MT2 attention fragout ----
kvA = PagedKVCache.for_model(cfg, n_pages=256, max_slots=44, max_ctx=64,
                             device="cuda", dtype=torch.bfloat16)
BA, LA = 40, 7
slotsA = [kvA.alloc_slot() for _ in range(BA)]
for s_ in slotsA:
    kvA.ensure(s_, LA)
kvA.k_pool.normal_(0, 0.3)
kvA.v_pool.normal_(0, 0.3)
qA = (torch.randn(BA, cfg.n_heads, 128, generator=g) * 0.3) \
    .bfloat16().cuda()
metaA = AttnMeta(mode="decode",
                 slot_ids=torch.tensor(slotsA, dtype=torch.int32,
                                       device="cuda"),
                 seq_lens=torch.full((BA,), LA, dtype=torch.int32,
                                     device="cuda"),
                 cu_q=torch.arange(BA + 1, dtype=torch.int32,
                                   device="cuda"),
                 logits_idx=None, max_q=1, max_kv=LA, window=0)
aA_std = hip.attention_decode(qA, kvA, 0, metaA)
aA_fr = hip.attention_decode(qA, kvA, 0, metaA, fragout=True)
KhA = cfg.n_heads * 128
print("mt2 attn-frag:", torch.allclose(
    defrag2(aA_fr, BA, KhA), aA_std.view(BA, KhA).float().cpu(),
    atol=5e-2, rtol=5e-2))

# ---- MT2 qkv-rope ----
model = LlamaModel(cfg, device="cuda", dtype=torch.bfloat16, seed=1)
lay = model.layers[0]
kvB = PagedKVCache.for_model(cfg, n_pages=256, max_slots=44, max_ctx=64,
                             device="cuda", dtype=torch.bfloat16)
BB = 40
slotsB = [kvB.alloc_slot() for _ in range(BB)]
LB = 5
for s_ in slotsB:
    kvB.ensure(s_, LB + 1)
posB = torch.full((BB,), LB, dtype=torch.int32, device="cuda")
slotB_t = torch.tensor(slotsB, dtype=torch.int32, device="cuda")
xB = (torch.randn(BB, cfg.hidden, generator=g) * 0.3).bfloat16().cuda()
xfB, _ = hip.fragify_sumsq(xB)
nl, nkl, d = cfg.n_heads, cfg.n_kv_heads, 128
NqB = (nl + 2 * nkl) * d
yB = hip.linear_qkv_rope(xfB, lay.wqkv_pk, NqB, None, kvB, 0, posB,
                         slotB_t, model.rope_cos, model.rope_sin,
                         nl, nkl, None, 0, 0.0, 0.0, K=cfg.hidden,
                         M_real=BB)
torch.cuda.synchronize()
qkv_ref = torch.nn.functional.linear(xB.float(), lay.wqkv.float())
qr, kr, vr = qkv_ref.split([nl * d, nkl * d, nkl * d], dim=-1)
co = model.rope_cos[LB].cpu()
si = model.rope_sin[LB].cpu()


def rope_ref(t, heads):
    t = t.view(BB, heads, d).cpu().clone()
    lo = t[..., :64].clone()
    hi = t[..., 64:].clone()
    t[..., :64] = lo * co - hi * si
    t[..., 64:] = hi * co + lo * si
    return t


q_exp = rope_ref(qr, nl)
print("mt2 rope-q:", torch.allclose(
    yB[:, :nl * d].view(BB, nl, d).float().cpu(), q_exp,
    atol=8e-2, rtol=8e-2))
k_exp = rope_ref(kr, nkl)
# read back pool at (slot, LB)
ps = kvB.page_size
ok = True
for i, s_ in enumerate(slotsB):
    pg = int(kvB.page_table[s_, LB // ps])
    krow = kvB.k_pool[0, pg, :, LB % ps].float().cpu()   # [KVH, D]
    vrow = kvB.v_pool[0, pg, :, LB % ps].float().cpu()
    if not torch.allclose(krow, k_exp[i], atol=8e-2, rtol=8e-2):
        ok = False
        print("  k mismatch row", i)
        break
    if not torch.allclose(vrow, vr[i].view(nkl, d).cpu(), atol=8e-2,
                          rtol=8e-2):
        ok = False
        print("  v mismatch row", i)
        break
print("mt2 qkv-rope pools:", ok)
