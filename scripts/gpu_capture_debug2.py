"""Test: does a trivial priming capture make dual-preset model captures
succeed?"""
import sys

sys.path.insert(0, "/root/repo")
import torch

from quoracle_amd.engine.engine import LocalEngine

dev = torch.device("cuda:0")
eng = LocalEngine(["mixtral-8x7b#h0", "llama3-8b#h0"], device=dev,
                  kv_gb_per_model=8.0)

# trivial priming capture
x = torch.zeros(64, device=dev)
s = torch.cuda.Stream(dev)
s.wait_stream(torch.cuda.current_stream(dev))
with torch.cuda.stream(s):
    x.add_(1)
torch.cuda.current_stream(dev).wait_stream(s)
torch.cuda.synchronize(dev)
gprime = torch.cuda.CUDAGraph()
try:
    with torch.cuda.graph(gprime, capture_error_mode="thread_local"):
        x.add_(1)
    print("priming capture: OK")
except Exception as exc:  # noqa: BLE001
    print(f"priming capture FAILED: {exc}")

eng.precapture_graphs()
for k, hm in eng.models.items():
    print(k, "graphs:", sorted(hm.graphs.graphs),
          "enabled:", hm.graphs.enabled)
