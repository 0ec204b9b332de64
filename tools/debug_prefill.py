"""Localize the MFMA prefill attention numerics error (GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from ollamamq_amd.engine.kvcache import PagedKVCache
from ollamamq_amd.ops.interface import AttnMeta
from ollamamq_amd.ops import reference as ref
from ollamamq_amd.ops import hip


def run_case(seq, Hq, KVH, label):
    D, L = 128, 1
    dev = "cuda"
    gc = PagedKVCache(L, KVH, D, page_size=16, n_pages=seq // 16 + 4,
                      max_slots=1, max_ctx=seq + 64, device=dev,
                      dtype=torch.bfloat16)
    cc = PagedKVCache(L, KVH, D, page_size=16, n_pages=seq // 16 + 4,
                      max_slots=1, max_ctx=seq + 64, device="cpu",
                      dtype=torch.float32)
    s = gc.alloc_slot(); cc.alloc_slot()
    gc.ensure(s, seq); cc.ensure(s, seq)
    cc.page_table.copy_(gc.page_table.cpu())
    g = torch.Generator().manual_seed(1)
    k = torch.randn(seq, KVH, D, generator=g).bfloat16().cuda()
    v = torch.randn(seq, KVH, D, generator=g).bfloat16().cuda()
    slot = torch.zeros(seq, dtype=torch.int32, device=dev)
    pos = torch.arange(seq, dtype=torch.int32, device=dev)
    hip.kv_append(gc, 0, k, v, slot, pos)
    ref.kv_append(cc, 0, k.float().cpu(), v.float().cpu(), slot.cpu(),
                  pos.cpu())
    q = torch.randn(seq, Hq, D, generator=g).bfloat16().cuda()
    meta = AttnMeta("prefill",
                    torch.zeros(1, dtype=torch.int32, device=dev),
                    torch.tensor([seq], dtype=torch.int32, device=dev),
                    torch.tensor([0, seq], dtype=torch.int32, device=dev),
                    None, seq, seq)
    meta_c = AttnMeta("prefill", torch.zeros(1, dtype=torch.int32),
                      torch.tensor([seq], dtype=torch.int32),
                      torch.tensor([0, seq], dtype=torch.int32),
                      None, seq, seq)
    out = hip.attention_prefill(q, gc, 0, meta)
    torch.cuda.synchronize()
    out_ref = ref.attention(q.float().cpu(), cc, 0, meta_c)
    err = (out.float().cpu() - out_ref).abs()     # [seq, Hq, D]
    per_q = err.amax(dim=(1, 2))
    print(f"{label}: max={err.max():.4f} "
          f"worst_q_rows={per_q.argsort(descending=True)[:5].tolist()} "
          f"per32={[round(per_q[i*32:(i+1)*32].max().item(),4) for i in range(min(8, seq//32))]}")
    per_h = err.amax(dim=(0, 2))
    print(f"   per-head max: {[round(x,4) for x in per_h.tolist()]}")


def main():
    hip.require()
    run_case(32, 4, 2, "seq=32 G=2 (single tile, single chunk)")
    run_case(64, 4, 2, "seq=64 G=2 (2 tiles, 1 chunk)")
    run_case(128, 4, 2, "seq=128 G=2")
    run_case(64, 8, 2, "seq=64 G=4")
    run_case(40, 4, 2, "seq=40 G=2 (ragged tile)")
    run_case(512, 32, 8, "seq=512 8B-shape")


def forensic():
    """Discriminate error classes for one small case."""
    seq, Hq, KVH, D, L = 32, 2, 2, 128, 1   # G=1!
    dev = "cuda"
    gc = PagedKVCache(L, KVH, D, page_size=16, n_pages=8, max_slots=1,
                      max_ctx=128, device=dev, dtype=torch.bfloat16)
    s = gc.alloc_slot()
    gc.ensure(s, seq)
    g = torch.Generator().manual_seed(2)
    k = torch.randn(seq, KVH, D, generator=g).bfloat16().cuda()
    v = torch.randn(seq, KVH, D, generator=g).bfloat16().cuda()
    hip.kv_append(gc, 0, k, v,
                  torch.zeros(seq, dtype=torch.int32, device=dev),
                  torch.arange(seq, dtype=torch.int32, device=dev))
    q = torch.randn(seq, Hq, D, generator=g).bfloat16().cuda()
    meta = AttnMeta("prefill",
                    torch.zeros(1, dtype=torch.int32, device=dev),
                    torch.tensor([seq], dtype=torch.int32, device=dev),
                    torch.tensor([0, seq], dtype=torch.int32, device=dev),
                    None, seq, seq)
    out = hip.attention_prefill(q, gc, 0, meta).float().cpu()
    torch.cuda.synchronize()
    qf, kf, vf = q.float().cpu(), k.float().cpu(), v.float().cpu()
    h = 0   # head 0 uses kv head 0
    sc = (qf[:, h] @ kf[:, 0].T) * (D ** -0.5)
    mask = torch.triu(torch.ones(seq, seq, dtype=torch.bool), 1)
    p = torch.softmax(sc.masked_fill(mask, float("-inf")), -1)
    ref = p @ vf[:, 0]
    o = out[:, h]
    print("G=1 err:", (o - ref).abs().max().item())
    # no-mask reference?
    p2 = torch.softmax(sc, -1)
    print("vs no-mask ref:", (o - p2 @ vf[:, 0]).abs().max().item())
    # unnormalized?
    e = torch.exp(sc.masked_fill(mask, float("-inf"))
                  - sc.masked_fill(mask, float("-inf")).amax(-1, True))
    print("vs unnormalized:", (o - e @ vf[:, 0]).abs().max().item())
    # d-permutation within rows? compare sorted rows
    so, sr = o.sort(dim=-1).values, ref.sort(dim=-1).values
    print("sorted-row err (d-permutation test):",
          (so - sr).abs().max().item())
    # q permutation? compare each out row to all ref rows
    d2 = ((o.unsqueeze(1) - ref.unsqueeze(0)) ** 2).mean(-1)
    print("best-match ref row for out rows 0..7:",
          d2.argmin(1)[:8].tolist())
    print("row 0 out[:6]:", o[0, :6].tolist())
    print("row 0 ref[:6]:", ref[0, :6].tolist())
    print("row 31 out[:6]:", o[31, :6].tolist())
    print("row 31 ref[:6]:", ref[31, :6].tolist())


if __name__ == "__main__":
    import sys as _s
    if "--forensic" in _s.argv:
        hip.require()
        forensic()
    else:
        main()
