import sys, math, time, torch
sys.path.insert(0, ".")
import bobrapet_amd._hipops as h

def bench(fn, iters=20, warm=4):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.monotonic()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.monotonic() - t0) / iters

B,S,Hq,Hkv,D = 4,2048,32,8,128
torch.manual_seed(0)
q = torch.randn(B,S,Hq,D, dtype=torch.bfloat16, device="cuda")
k = torch.randn(B,S,Hkv,D, dtype=torch.bfloat16, device="cuda")
v = torch.randn(B,S,Hkv,D, dtype=torch.bfloat16, device="cuda")
sc = 1/math.sqrt(D)
for causal in (False, True):
    times = {}
    for var in (1, 3, 7):
        times[var] = bench(lambda: h.attn_prefill_variant(var, q, k, v, sc, causal)) * 1e3
    print(f"causal={int(causal)}: stage={times[1]:.3f}ms  +qk/sm={times[3]:.3f}ms  full={times[7]:.3f}ms"
          f"  (qk/sm adds {times[3]-times[1]:.3f}, pv adds {times[7]-times[3]:.3f})")
