"""Single-GPU Llama-3-70B bf16: 131 GB of weights resident in 288 GB HBM3E.
Prefill + hipGraph-replayed decode on one MI355X — no tensor parallelism."""
import sys
import time

import torch

sys.path.insert(0, ".")
from bobrapet_amd.models.llama import LlamaModel  # noqa: E402

t0 = time.perf_counter()
m = LlamaModel("llama-3-70b", device="cuda")
torch.cuda.synchronize()
print(f"init {time.perf_counter()-t0:.1f}s  params {m.param_bytes/2**30:.1f} GiB")

ids = torch.randint(0, m.cfg.vocab_size, (1, 512), device="cuda")
t0 = time.perf_counter()
logits = m.prefill(ids, fill_cache=True)
torch.cuda.synchronize()
pf = time.perf_counter() - t0
print(f"prefill b1 s512: {pf*1e3:.0f} ms  ({512/pf:.0f} tok/s)")

nxt = torch.randint(0, m.cfg.vocab_size, (1,), device="cuda")
for _ in range(3):
    nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize()
t0 = time.perf_counter()
n = 30
for _ in range(n):
    nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / n * 1e3
print(f"decode b1: {ms:.2f} ms/token  {1e3/ms:.1f} tok/s")
print(f"HBM allocated {torch.cuda.memory_allocated()/2**30:.1f} GiB / "
      f"reserved {torch.cuda.memory_reserved()/2**30:.1f} GiB")
print("OK")
