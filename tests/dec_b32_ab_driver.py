"""A/B: b32 graphed decode with v3 two-pass (default) vs BOBRA_DEC_V1 single-pass.
BOBRA_DEC_V1 must be set before the first decode (graph capture bakes the path),
so each arm runs in a subprocess."""
import os
import subprocess
import sys

CHILD = r"""
import sys, time, torch
sys.path.insert(0, ".")
from bobrapet_amd.models.llama import LlamaModel
B = int(sys.argv[1])
m = LlamaModel("llama-3-8b", device="cuda")
ids = torch.randint(0, m.cfg.vocab_size, (B, 512), device="cuda")
m.prefill(ids, fill_cache=True)
nxt = torch.randint(0, m.cfg.vocab_size, (B,), device="cuda")
for _ in range(3):
    nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize()
t0 = time.perf_counter()
n = 40
for _ in range(n):
    nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize()
ms = (time.perf_counter() - t0) / n * 1e3
print(f"B={B} {ms:.2f} ms/step {B/ms*1e3:.0f} tok/s", flush=True)
"""

for B in (1, 32):
    for tag, env in (("v3", {}), ("v1", {"BOBRA_DEC_V1": "1"})):
        e = dict(os.environ, **env)
        r = subprocess.run(
            [sys.executable, "-c", CHILD, str(B)], env=e, capture_output=True, text=True
        )
        line = (r.stdout.strip().splitlines() or ["<no output>"])[-1]
        print(f"{tag}: {line}")
        if r.returncode != 0:
            print(r.stderr[-1500:])
