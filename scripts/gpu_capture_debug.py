"""Diagnose the dual-preset first-model hipGraph capture failure: rerun
the capture body manually and print the real traceback."""
import sys
import traceback

sys.path.insert(0, "/root/repo")
import torch

from quoracle_amd.engine.engine import LocalEngine
from quoracle_amd.models.llama import ForwardBatch

eng = LocalEngine(["mixtral-8x7b#h0", "llama3-8b#h0"],
                  device=torch.device("cuda:0"), kv_gb_per_model=8.0)
g = eng.models["mixtral-8x7b#h0"].graphs
dev = g.device
bucket = 1
bufs = {
    "tokens": torch.zeros(bucket, dtype=torch.int32, device=dev),
    "positions": torch.zeros(bucket, dtype=torch.int32, device=dev),
    "slots": torch.full((bucket,), g.scratch_block * g.kv.block_size,
                        dtype=torch.int32, device=dev),
    "block_tables": torch.full((bucket, g.maxb), g.scratch_block,
                               dtype=torch.int32, device=dev),
    "ctx_lens": torch.ones(bucket, dtype=torch.int32, device=dev),
}
rows = torch.arange(bucket, dtype=torch.long, device=dev)


def fwd():
    b = ForwardBatch(tokens=bufs["tokens"], positions=bufs["positions"],
                     slots=bufs["slots"],
                     block_tables=bufs["block_tables"], n_decode=bucket,
                     ctx_lens=bufs["ctx_lens"], max_ctx=1 << 30)
    h = g.model.forward(b, g.kv)
    return g.model.compute_logits(h, rows)


s = torch.cuda.Stream(dev)
s.wait_stream(torch.cuda.current_stream(dev))
with torch.cuda.stream(s):
    for _ in range(2):
        fwd()
torch.cuda.current_stream(dev).wait_stream(s)
torch.cuda.synchronize(dev)
gr = torch.cuda.CUDAGraph()
try:
    with torch.cuda.graph(gr, capture_error_mode="thread_local"):
        fwd()
    print("manual capture OK")
except Exception:
    traceback.print_exc()
