"""70B single-GPU serving snapshot at b1/b4/b8 (bf16, graphed decode)."""
import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bobrapet_amd.models.llama import LlamaModel

m = LlamaModel("llama-3-70b", device="cuda")
torch.cuda.synchronize()
print(f"params {m.param_bytes/2**30:.1f} GiB", flush=True)
for B in (1, 4, 8):
    ids = torch.randint(0, m.cfg.vocab_size, (B, 512), device="cuda")
    t0 = time.perf_counter()
    m.prefill(ids, fill_cache=True)
    torch.cuda.synchronize()
    pf = (time.perf_counter() - t0) * 1e3
    nxt = torch.randint(0, m.cfg.vocab_size, (B,), device="cuda")
    for _ in range(5):
        nxt = m.decode_step_graphed(nxt).argmax(-1)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(40):
        nxt = m.decode_step_graphed(nxt).argmax(-1)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 40 * 1e3
    print(f"70B b{B}: prefill(s512) {pf:7.1f} ms   decode {ms:6.2f} ms/step = "
          f"{B/ms*1e3:6.0f} tok/s", flush=True)
print(f"HBM reserved {torch.cuda.memory_reserved()/2**30:.1f} GiB", flush=True)
