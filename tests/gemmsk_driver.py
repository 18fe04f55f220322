"""Skinny-M GEMM (decode batch 9..32): numerics for all epilogues at
several M, then per-shape timing on the 8B decode projections at M=32.
GPU box:  python tests/gemmsk_driver.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops

hip = ops._try_load()
assert hip is not None, ops._load_error


def timed(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def rel(got, ref):
    return ((got.float() - ref).abs().max() / ref.abs().max()).item()


def main():
    torch.manual_seed(4)
    dev = "cuda"
    # ---- numerics ----
    for m in (9, 16, 32):
        k, n = 4096, 1536
        a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        r = (torch.randn(m, n, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        stat = ops.rowsumsq(a)
        ref0 = torch.matmul(a.float(), b.float().t())
        scale = torch.rsqrt(stat.float() / k + 1e-5)[:, None]
        (g0,) = hip.gemmsk(a, b, 0, None, stat, 1.0 / k, 1e-5)
        print(f"M{m}: epi0 relerr={rel(g0, ref0 * scale):.3e}", flush=True)
        (g1,) = hip.gemmsk(a, b, 1, None, stat, 1.0 / k, 1e-5)
        gg, uu = (ref0 * scale)[:, 0::2], (ref0 * scale)[:, 1::2]
        ref1 = torch.nn.functional.silu(gg) * uu
        print(f"M{m}: epi1 swiglu relerr={rel(g1, ref1):.3e}", flush=True)
        g2, so = hip.gemmsk(a, b, 2, r, None, 0, 0)
        ref2 = ref0 + r.float()
        print(f"M{m}: epi2 resid relerr={rel(g2, ref2):.3e} "
              f"stat relerr={rel(so, ref2.pow(2).sum(-1)):.3e}", flush=True)
        outs = [hip.gemmsk(a, b, 0, None, stat, 1.0 / k, 1e-5)[0] for _ in range(3)]
        print(f"M{m}: self-stable={all(torch.equal(outs[0], o) for o in outs[1:])}",
              flush=True)

    # ---- perf (8B decode shapes, M=32) ----
    M = 32
    shapes = [("qkv", 6144, 4096, 0), ("o", 4096, 4096, 2),
              ("gateup", 28672, 4096, 1), ("down", 4096, 14336, 2)]
    total = 0.0
    for name, n, k, epi in shapes:
        a = (torch.randn(M, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.05).contiguous()
        r = (torch.randn(M, n if epi == 2 else 64, dtype=torch.bfloat16,
                         device=dev) * 0.3).contiguous()
        stat = ops.rowsumsq(a)
        if epi == 2:
            us = timed(lambda: hip.gemmsk(a, b, 2, r, None, 0, 0))
        else:
            us = timed(lambda: hip.gemmsk(a, b, epi, None, stat, 1.0 / k, 1e-5))
        mb = n * k * 2 / 1e6
        total += us
        print(f"{name:7s} N{n} K{k} epi{epi}: {us:6.1f} us "
              f"({mb/1e3/(us*1e-6):5.2f} TB/s weights)", flush=True)
    print(f"layer total {total:6.1f} us -> est step "
          f"{total*32/1e3:5.2f} ms GEMM-only", flush=True)


if __name__ == "__main__":
    main()
