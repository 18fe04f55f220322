import sys, os, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bobrapet_amd.models.llama import LlamaModel
m = LlamaModel("llama-3-8b", device="cuda")
ids = torch.randint(0, m.cfg.vocab_size, (4, 2048), device="cuda")
for _ in range(3):
    m.prefill(ids)
torch.cuda.synchronize()
print("profiled")
