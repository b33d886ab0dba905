"""Scale repro: llama3-8b graph decode at bench-like shapes."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
if os.environ.get("QUORACLE_BLAS") == "rocblas":
    torch.backends.cuda.preferred_blas_library("cublas")
from quoracle_amd.models import LlamaModel
from quoracle_amd.models.llama import ForwardBatch
from quoracle_amd.engine.graphs import DecodeGraphs

dev = torch.device("cuda:0")
stage = sys.argv[1] if len(sys.argv) > 1 else "single"

if stage == "single":
    model = LlamaModel("llama3-8b#x", dev)
    print("model up", flush=True)
    NBLK = 1024
    kv = model.new_kv_cache(NBLK, 16)
    graphs = DecodeGraphs(model, kv, dev, max_blocks_per_seq=512, scratch_block=0)
    CTX = 7000
    nb = (CTX + 15) // 16 + 2
    blocks = list(range(1, nb + 1))
    bt = torch.zeros((1, 512), dtype=torch.int32, device=dev)
    bt[0, :nb] = torch.tensor(blocks, dtype=torch.int32)
    # chunked prefill 2048 at a time
    for start in range(0, CTX, 2048):
        n = min(2048, CTX - start)
        toks = (torch.arange(n, dtype=torch.int32, device=dev) % 511)
        pos = torch.arange(start, start + n, dtype=torch.int32, device=dev)
        slots = torch.tensor([blocks[p // 16] * 16 + p % 16
                              for p in range(start, start + n)],
                             dtype=torch.int32, device=dev)
        ntiles = (n + 15) // 16
        t0 = torch.arange(ntiles, dtype=torch.int32, device=dev) * 16
        qn = torch.clamp(torch.full_like(t0, n) - t0, max=16)
        b = ForwardBatch(tokens=toks, positions=pos, slots=slots,
                         block_tables=bt, n_decode=0, tile_q0=t0, tile_qn=qn,
                         tile_seq=torch.zeros_like(t0), tile_pos0=t0 + start)
        model.forward(b, kv)
        torch.cuda.synchronize()
        print(f"prefill {start}+{n} ok", flush=True)

    pos = CTX
    for i in range(40):
        print(f"replay {i}", flush=True) if i < 5 else None
        tok, slot, ctx = [int(7 + i)], [blocks[pos // 16] * 16 + pos % 16], [pos + 1]
        logits = graphs.run(tok, [pos], slot, [blocks], ctx)
        assert logits is not None, "graphs disabled"
        if i == 0:
            torch.cuda.synchronize(); print("first replay ok", flush=True)
        nxt = int(logits[0].argmax())
        pos += 1
    torch.cuda.synchronize()
    print("40 graph decode steps ok", flush=True)
    # eager check of final step
    b = ForwardBatch(
        tokens=torch.tensor([7 + 39], dtype=torch.int32, device=dev),
        positions=torch.tensor([pos - 1], dtype=torch.int32, device=dev),
        slots=torch.tensor([blocks[(pos - 1) // 16] * 16 + (pos - 1) % 16],
                           dtype=torch.int32, device=dev),
        block_tables=bt[:, :512], n_decode=1,
        ctx_lens=torch.tensor([pos], dtype=torch.int32, device=dev),
        max_ctx=pos)
    h = model.forward(b, kv)
    el = model.compute_logits(h, torch.tensor([0], device=dev))
    rel = (el.float() - logits.float()).norm() / el.float().norm()
    print("rel eager-vs-graph:", rel.item(), flush=True)
    print("PASS single", flush=True)

elif stage == "engine":
    from quoracle_amd.engine.engine import LocalEngine
    from quoracle_amd.engine.api import GenerateRequest
    eng = LocalEngine([f"llama3-8b#m{i}" for i in range(3)],
                      device=dev, kv_gb_per_model=4.0)
    print("engine up", flush=True)
    msgs = [{"role": "system", "content": "x" * 7000},
            {"role": "user", "content": "assess the task"}]
    import threading
    results = []
    reqs = [GenerateRequest(model_key=f"llama3-8b#m{i % 3}", messages=msgs,
                            temperature=0.8, max_tokens=1024, seed=i,
                            action_grammar=True, session_id=f"s{i}")
            for i in range(6)]
    done = []
    ev = threading.Event()
    for r in reqs:
        eng._submit(r, ("cb", lambda res: (done.append(res),
                                           len(done) == 6 and ev.set())))
    eng.precapture_graphs((1, 2))
    import time
    t0 = time.time(); last = t0
    while not ev.is_set():
        eng.step()
        now = time.time()
        if now - last > 10:
            last = now
            print(f"hb t={now - t0:.0f}s stats={eng.stats}", flush=True)
        if now - t0 > 280:
            print("TIMEOUT", flush=True); sys.exit(2)
    for r in done:
        print(r.model_key, r.error, r.output_tokens, flush=True)
    assert all(r.ok for r in done)
    print("PASS engine", flush=True)
