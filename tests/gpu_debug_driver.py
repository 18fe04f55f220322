import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, math
from bobrapet_amd import ops

torch.manual_seed(0)

def check(tag, b, h, hkv, s, causal):
    D = 128
    q = torch.randn(b, h, s, D, dtype=torch.bfloat16, device="cuda")
    k = torch.randn(b, hkv, s, D, dtype=torch.bfloat16, device="cuda")
    v = torch.randn(b, hkv, s, D, dtype=torch.bfloat16, device="cuda")
    sc = 1.0 / math.sqrt(D)
    got = ops.attn_prefill(q, k, v, sc, causal)
    ref = ops.attn_ref(q, k, v, sc, causal)
    err = (got.float() - ref.float()).abs()
    print(f"{tag}: max={err.max().item():.4f}")
    if err.max().item() > 0.05:
        # error structure: per q-row max, per d-col max
        per_q = err[0,0].max(dim=-1).values
        per_d = err[0,0].max(dim=0).values
        bad_q = (per_q > 0.05).nonzero().flatten().tolist()
        bad_d = (per_d > 0.05).nonzero().flatten().tolist()
        print("  bad q rows:", bad_q[:20], "..." if len(bad_q)>20 else "", f"({len(bad_q)} total)")
        print("  bad d cols:", bad_d[:20], "..." if len(bad_d)>20 else "", f"({len(bad_d)} total)")

check("S=64 nc ", 1, 1, 1, 64, False)
check("S=128 nc", 1, 1, 1, 128, False)
check("S=128 c ", 1, 1, 1, 128, True)
check("S=256 nc", 1, 1, 1, 256, False)
