"""Isolate the hipGraph decode crash: tiny model, eager vs graph logits."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from quoracle_amd.models import LlamaModel
from quoracle_amd.models.llama import ForwardBatch
from quoracle_amd.engine.graphs import DecodeGraphs

dev = torch.device("cuda:0")
model = LlamaModel("tiny#g", dev)
kv = model.new_kv_cache(64, 16)
mgr_scratch = 0
graphs = DecodeGraphs(model, kv, dev, max_blocks_per_seq=8, scratch_block=0)

# build two sequences with some cached context via eager prefill
def prefill(seq_blocks, n, seq_row, bt):
    toks = (torch.arange(n, dtype=torch.int32, device=dev) % 512)
    pos = torch.arange(n, dtype=torch.int32, device=dev)
    slots = torch.tensor([seq_blocks[p // 16] * 16 + p % 16 for p in range(n)],
                         dtype=torch.int32, device=dev)
    ntiles = (n + 15) // 16
    t0 = torch.arange(ntiles, dtype=torch.int32, device=dev) * 16
    qn = torch.clamp(torch.full_like(t0, n) - t0, max=16)
    b = ForwardBatch(tokens=toks, positions=pos, slots=slots,
                     block_tables=bt, n_decode=0, tile_q0=t0, tile_qn=qn,
                     tile_seq=torch.full_like(t0, seq_row), tile_pos0=t0)
    model.forward(b, kv)

bt = torch.tensor([[1, 2, 3, 0, 0, 0, 0, 0],
                   [4, 5, 6, 7, 0, 0, 0, 0]], dtype=torch.int32, device=dev)
prefill([1, 2, 3], 40, 0, bt)
prefill([4, 5, 6, 7], 60, 1, bt)
torch.cuda.synchronize()
print("prefill ok", flush=True)

tokens, positions, slots = [7, 9], [40, 60], [2 * 16 + 8, 6 * 16 + 12]
ctx = [41, 61]
# eager decode
b = ForwardBatch(tokens=torch.tensor(tokens, dtype=torch.int32, device=dev),
                 positions=torch.tensor(positions, dtype=torch.int32, device=dev),
                 slots=torch.tensor(slots, dtype=torch.int32, device=dev),
                 block_tables=bt, n_decode=2,
                 ctx_lens=torch.tensor(ctx, dtype=torch.int32, device=dev),
                 max_ctx=1 << 30)
hidden = model.forward(b, kv)
eager_logits = model.compute_logits(hidden, torch.tensor([0, 1], device=dev))
torch.cuda.synchronize()
print("eager decode ok", flush=True)

logits = graphs.run(tokens, positions, slots, [[1, 2, 3], [4, 5, 6, 7]], ctx)
torch.cuda.synchronize()
print("graph replay ok; enabled:", graphs.enabled, flush=True)
if logits is None:
    print("GRAPHS DISABLED (capture failed)"); sys.exit(1)
rel = (logits.float() - eager_logits.float()).norm() / eager_logits.float().norm()
print("rel diff eager-vs-graph:", rel.item(), flush=True)
# replay again with different tokens (stability)
logits2 = graphs.run([3, 4], [41, 61], [2 * 16 + 9, 6 * 16 + 13], [[1, 2, 3], [4, 5, 6, 7]], [42, 62])
torch.cuda.synchronize()
print("second replay ok", flush=True)
assert rel.item() < 0.25, rel.item()
print("PASS", flush=True)
