import sys, time, torch
sys.path.insert(0, ".")
from bobrapet_amd.models.llama import LlamaModel
m = LlamaModel("llama-3-8b", device="cuda")
ids = torch.randint(0, m.cfg.vocab_size, (1, 512), device="cuda")
m.prefill(ids, fill_cache=True)
nxt = torch.randint(0, m.cfg.vocab_size, (1,), device="cuda")
for _ in range(10): nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(200): nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize(); ms = (time.perf_counter() - t0) / 200 * 1e3
print(f"8B decode b1: {ms:.2f} ms/token = {1e3/ms:.1f} tok/s", flush=True)
