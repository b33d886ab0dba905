"""Micro-bench: MFMA vs VALU prefill attention at bench-like shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quoracle_amd import ops

dev = torch.device("cuda:0")
ops.ext()
Hq, Hkv, D, BS = 32, 8, 128, 16
scale = D ** -0.5
cached, new = 5000, 2048
total = cached + new
nb = (total + BS - 1) // BS
kcache = torch.randn(nb + 1, Hkv, BS, D, device=dev, dtype=torch.bfloat16)
vcache = torch.randn_like(kcache)
tables = torch.arange(1, nb + 1, dtype=torch.int32, device=dev).unsqueeze(0)
q = torch.randn(new, Hq, D, device=dev, dtype=torch.bfloat16)
out = torch.empty_like(q)
ntiles = (new + 15) // 16
t0 = torch.arange(ntiles, dtype=torch.int32, device=dev) * 16
qn = torch.clamp(torch.full_like(t0, new) - t0, max=16)
tseq = torch.zeros_like(t0)
tpos = t0 + cached

def timeit(fn, n=20):
    fn(); torch.cuda.synchronize()
    t = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / n * 1e3

ms_valu = timeit(lambda: ops.ext().paged_attn_prefill(
    out, q, kcache, vcache, tables, t0, qn, tseq, tpos, scale))
out_valu = out.clone()
ms_mfma = timeit(lambda: ops.ext().paged_attn_prefill_mfma(
    out, q, kcache, vcache, tables, t0, qn, tseq, tpos, scale))
out_mfma = out.clone()
rel = (out_mfma.float() - out_valu.float()).norm() / out_valu.float().norm()
# 32-row tile metadata for the 8-wave variant
nt32 = (new + 31) // 32
t32 = torch.arange(nt32, dtype=torch.int32, device=dev) * 32
qn32 = torch.clamp(torch.full_like(t32, new) - t32, max=32)
ts32 = torch.zeros_like(t32)
tp32 = t32 + cached
ms_m32 = timeit(lambda: ops.ext().paged_attn_prefill_mfma32(
    out, q, kcache, vcache, tables, t32, qn32, ts32, tp32, scale))
rel32 = (out.float() - out_mfma.float()).norm() / out_mfma.float().norm()
flops = 2 * 2 * new * (cached + new / 2) * D * Hq
print(f"VALU prefill:   {ms_valu:.3f} ms ({flops/ms_valu/1e9:.1f} TFLOP/s)")
print(f"MFMA16 prefill: {ms_mfma:.3f} ms ({flops/ms_mfma/1e9:.1f} TFLOP/s)  "
      f"speedup {ms_valu/ms_mfma:.1f}x  rel-vs-valu {rel:.4f}")
print(f"MFMA32 prefill: {ms_m32:.3f} ms ({flops/ms_m32/1e9:.1f} TFLOP/s)  "
      f"vs16 {ms_mfma/ms_m32:.2f}x  rel-vs-16 {rel32:.4f}")

# --- decode path at bench-like shapes (B agents x 1 token, long ctx) -------
B = 8
ctx_len = 7000
nb2 = (ctx_len + BS - 1) // BS + 1
tables2 = torch.stack([torch.arange(1, nb2, dtype=torch.int32, device=dev)
                       for _ in range(B)])
qd = torch.randn(B, Hq, D, device=dev, dtype=torch.bfloat16)
outd = torch.empty_like(qd)
ctxs = torch.full((B,), ctx_len, dtype=torch.int32, device=dev)
NSd = min(32, max(2, ctx_len // 256))
pm = torch.empty((B, Hq, NSd), dtype=torch.float32, device=dev)
pl = torch.empty_like(pm)
pa = torch.empty((B, Hq, NSd, D), dtype=torch.float32, device=dev)
ms_dec = timeit(lambda: ops.ext().paged_attn_decode_split(
    outd, qd, kcache, vcache, tables2, ctxs, scale, pm, pl, pa), n=50)
kv_bytes = 2 * B * ctx_len * Hkv * D * 2      # K+V bf16 read per call
print(f"decode_split B={B} ctx={ctx_len} NS={NSd}: {ms_dec*1e3:.1f} us "
      f"({kv_bytes/ms_dec/1e9:.2f} TB/s of {8} TB/s HBM peak)")

# --- experimental kernels (validate with tests/test_ops_experimental.py
# BEFORE trusting these numbers) --------------------------------------------
if hasattr(ops.ext(), "paged_attn_prefill_mfma64"):
    nt64 = (new + 63) // 64
    t64 = torch.arange(nt64, dtype=torch.int32, device=dev) * 64
    qn64 = torch.clamp(torch.full_like(t64, new) - t64, max=64)
    ts64 = torch.zeros_like(t64)
    tp64 = t64 + cached
    ms_m64 = timeit(lambda: ops.ext().paged_attn_prefill_mfma64(
        out, q, kcache, vcache, tables, t64, qn64, ts64, tp64, scale))
    rel64 = (out.float() - out_mfma.float()).norm() / out_mfma.float().norm()
    print(f"MFMA64 prefill (EXPERIMENTAL): {ms_m64:.3f} ms "
          f"({flops/ms_m64/1e9:.1f} TFLOP/s)  rel-vs-16 {rel64:.4f}")
if hasattr(ops.ext(), "paged_attn_decode_split2"):
    outd2 = torch.empty_like(qd)
    ms_dec2 = timeit(lambda: ops.ext().paged_attn_decode_split2(
        outd2, qd, kcache, vcache, tables2, ctxs, scale, pm, pl, pa), n=50)
    reld = (outd2.float() - outd.float()).norm() / outd.float().norm()
    print(f"decode_split2 (EXPERIMENTAL): {ms_dec2*1e3:.1f} us "
          f"({kv_bytes/ms_dec2/1e9:.2f} TB/s)  rel-vs-v1 {reld:.4f}")
if hasattr(ops.ext(), "paged_attn_decode_mfma"):
    outdm = torch.empty_like(qd)
    ms_decm = timeit(lambda: ops.ext().paged_attn_decode_mfma(
        outdm, qd, kcache, vcache, tables2, ctxs, scale, pm, pl, pa), n=50)
    reldm = (outdm.float() - outd.float()).norm() / outd.float().norm()
    print(f"decode_mfma: {ms_decm*1e3:.1f} us "
          f"({kv_bytes/ms_decm/1e9:.2f} TB/s)  rel-vs-v1 {reldm:.4f}")
    # decode-at-low-batch case (single agent turn)
    for B1 in (1, 2):
        q1 = qd[:B1].contiguous(); o1 = torch.empty_like(q1)
        t1_ = tables2[:B1].contiguous(); c1 = ctxs[:B1].contiguous()
        m1 = timeit(lambda: ops.ext().paged_attn_decode_mfma(
            o1, q1, kcache, vcache, t1_, c1, scale,
            pm[:B1].contiguous(), pl[:B1].contiguous(),
            pa[:B1].contiguous()), n=50)
        kb1 = 2 * B1 * ctx_len * Hkv * D * 2
        print(f"decode_mfma B={B1}: {m1*1e3:.1f} us ({kb1/m1/1e9:.2f} TB/s)")
if hasattr(ops.ext(), "paged_attn_prefill_t12"):
    nt128 = (new + 127) // 128
    t128 = torch.arange(nt128, dtype=torch.int32, device=dev) * 128
    qn128 = torch.clamp(torch.full_like(t128, new) - t128, max=128)
    ts128 = torch.zeros_like(t128)
    tp128 = t128 + cached
    ms_t12 = timeit(lambda: ops.ext().paged_attn_prefill_t12(
        out, q, kcache, vcache, tables, t128, qn128, ts128, tp128, scale))
    rel12 = (out.float() - out_mfma.float()).norm() / out_mfma.float().norm()
    print(f"T12 prefill (EXPERIMENTAL): {ms_t12:.3f} ms "
          f"({flops/ms_t12/1e9:.1f} TFLOP/s)  rel-vs-16 {rel12:.4f}")
# --- small-chunk split path (grammar forced-runs: 23 new tokens over a
# long cached context) ------------------------------------------------------
sc_new = 23
qsc = torch.randn(sc_new, Hq, D, device=dev, dtype=torch.bfloat16)
osc = torch.empty_like(qsc)
nt_sc = (sc_new + 15) // 16
t0s = torch.arange(nt_sc, dtype=torch.int32, device=dev) * 16
qns = torch.clamp(torch.full_like(t0s, sc_new) - t0s, max=16)
tss = torch.zeros_like(t0s)
tps = t0s + cached
NSs = 16
pms = torch.empty((nt_sc, Hq, NSs, 16), dtype=torch.float32, device=dev)
pls = torch.empty_like(pms)
pas = torch.empty((nt_sc, Hq, NSs, 16, D), dtype=torch.float32, device=dev)
ms_spl = timeit(lambda: ops.ext().paged_attn_prefill_mfma_split(
    osc, qsc, kcache, vcache, tables, t0s, qns, tss, tps, scale,
    pms, pls, pas), n=50)
o_spl = osc.clone()
print(f"mfma_split small-chunk ({sc_new} tok / {cached} ctx, NS={NSs}): "
      f"{ms_spl*1e3:.1f} us")
if hasattr(ops.ext(), "paged_attn_prefill_t12_split"):
    NS8 = NSs * 8
    pm8 = torch.empty((nt_sc, Hq, NS8, 16), dtype=torch.float32, device=dev)
    pl8 = torch.empty_like(pm8)
    pa8 = torch.empty((nt_sc, Hq, NS8, 16, D), dtype=torch.float32,
                      device=dev)
    ms_t12s = timeit(lambda: ops.ext().paged_attn_prefill_t12_split(
        osc, qsc, kcache, vcache, tables, t0s, qns, tss, tps, scale,
        pm8, pl8, pa8), n=50)
    rel_s = (osc.float() - o_spl.float()).norm() / o_spl.float().norm()
    print(f"t12_split small-chunk (EXPERIMENTAL): {ms_t12s*1e3:.1f} us "
          f"({ms_spl/ms_t12s:.2f}x)  rel-vs-mfma_split {rel_s:.4f}")
if hasattr(ops.ext(), "paged_attn_prefill_t12w"):
    ms_t12w = timeit(lambda: ops.ext().paged_attn_prefill_t12w(
        out, q, kcache, vcache, tables, t128, qn128, ts128, tp128, scale))
    rel12w = (out.float() - out_mfma.float()).norm() / out_mfma.float().norm()
    print(f"T12W prefill (EXPERIMENTAL 32x32): {ms_t12w:.3f} ms "
          f"({flops/ms_t12w/1e9:.1f} TFLOP/s)  rel-vs-16 {rel12w:.4f}")
# --- decode NS sensitivity (pick the split-count formula) ------------------
if hasattr(ops.ext(), "paged_attn_decode_mfma"):
    for NSx in (8, 16, 27, 40, 56):
        pmx = torch.empty((B, Hq, NSx), dtype=torch.float32, device=dev)
        plx = torch.empty_like(pmx)
        pax = torch.empty((B, Hq, NSx, D), dtype=torch.float32, device=dev)
        msx = timeit(lambda: ops.ext().paged_attn_decode_mfma(
            outd, qd, kcache, vcache, tables2, ctxs, scale,
            pmx, plx, pax), n=50)
        m1x = timeit(lambda: ops.ext().paged_attn_decode_mfma(
            outd[:1].contiguous(), qd[:1].contiguous(), kcache, vcache,
            tables2[:1].contiguous(), ctxs[:1].contiguous(), scale,
            pmx[:1].contiguous(), plx[:1].contiguous(),
            pax[:1].contiguous()), n=50)
        print(f"decode_mfma NS={NSx}: B=8 {msx*1e3:.1f} us "
              f"({kv_bytes/msx/1e9:.2f} TB/s)  B=1 {m1x*1e3:.1f} us")
# --- context-scaling sweep (QUORACLE_ATTN_SWEEP=1) -------------------------
if os.environ.get("QUORACLE_ATTN_SWEEP"):
    for ctxs_len in (2000, 7000, 16000, 32000):
        nbS = (ctxs_len + BS - 1) // BS + 1
        kcS = torch.randn(nbS, Hkv, BS, D, device=dev, dtype=torch.bfloat16)
        vcS = torch.randn_like(kcS)
        tabS = torch.stack([torch.arange(1, nbS, dtype=torch.int32,
                                         device=dev) for _ in range(8)])
        qS = torch.randn(8, Hq, D, device=dev, dtype=torch.bfloat16)
        oS = torch.empty_like(qS)
        cS = torch.full((8,), ctxs_len, dtype=torch.int32, device=dev)
        NSS = min(32, max(2, ctxs_len // 256))
        pmS = torch.empty((8, Hq, NSS), dtype=torch.float32, device=dev)
        plS = torch.empty_like(pmS)
        paS = torch.empty((8, Hq, NSS, D), dtype=torch.float32, device=dev)
        msS = timeit(lambda: ops.ext().paged_attn_decode_mfma(
            oS, qS, kcS, vcS, tabS, cS, scale, pmS, plS, paS), n=30)
        kvb = 2 * 8 * ctxs_len * Hkv * D * 2
        # T12 prefill: 1024 new tokens over this cached context
        newS = 1024
        qP = torch.randn(newS, Hq, D, device=dev, dtype=torch.bfloat16)
        oP = torch.empty_like(qP)
        ntS = (newS + 127) // 128
        t0S = torch.arange(ntS, dtype=torch.int32, device=dev) * 128
        qnS = torch.clamp(torch.full_like(t0S, newS) - t0S, max=128)
        tsS = torch.zeros_like(t0S)
        tpS = t0S + ctxs_len - newS
        msP = timeit(lambda: ops.ext().paged_attn_prefill_t12(
            oP, qP, kcS, vcS, tabS[:1].contiguous(), t0S, qnS, tsS, tpS,
            scale), n=10)
        flP = 2 * 2 * newS * (ctxs_len - newS / 2) * D * Hq
        print(f"ctx={ctxs_len:6d}: decode_mfma B=8 {msS*1e3:7.1f} us "
              f"({kvb/msS/1e9:.2f} TB/s)   t12 prefill 1024-tok chunk "
              f"{msP:7.3f} ms ({flP/msP/1e9:.0f} TF)")
