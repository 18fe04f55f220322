import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bobrapet_amd.models.llama import LlamaModel
m = LlamaModel("llama-3-8b", device="cuda")
ids = torch.randint(0, m.cfg.vocab_size, (32, 512), device="cuda")
m.prefill(ids, fill_cache=True)
nxt = torch.randint(0, m.cfg.vocab_size, (32,), device="cuda")
for _ in range(12):
    nxt = m.decode_step_graphed(nxt).argmax(-1)
torch.cuda.synchronize()
print("done")
