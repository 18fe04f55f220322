"""gemm256w (32x32x16 MFMA) vs gemm256b (16x16x32): numerics for all
three epilogues, then interleaved perf on the Llama-3-8B prefill shapes.
GPU box:  python tests/gemm256w_driver.py [M]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops

hip = ops._try_load()
assert hip is not None, ops._load_error


def timed(fn, iters=10):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def relerr(got, ref):
    return ((got.float() - ref).abs().max() / ref.abs().max()).item()


def main():
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    torch.manual_seed(9)
    dev = "cuda"
    # ---- numerics (M with tail, plus aligned) ----
    for m in (300, 512):
        k, n = 4096, 1536
        a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        r = (torch.randn(m, n, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        stat = ops.rowsumsq(a)
        ref0 = torch.matmul(a.float(), b.float().t())
        scale = torch.rsqrt(stat.float() / k + 1e-5)[:, None]
        print(f"M{m}: epi0 relerr={relerr(hip.gemm256_w(a, b, 0, None, None, 0, 0), ref0):.3e}",
              flush=True)
        ref0s = ref0 * scale
        print(f"M{m}: epi0+stat relerr="
              f"{relerr(hip.gemm256_w(a, b, 0, None, stat, 1.0 / k, 1e-5), ref0s):.3e}",
              flush=True)
        g, u = ref0s[:, 0::2], ref0s[:, 1::2]
        ref1 = torch.nn.functional.silu(g) * u
        print(f"M{m}: epi1 swiglu relerr="
              f"{relerr(hip.gemm256_w(a, b, 1, None, stat, 1.0 / k, 1e-5), ref1):.3e}",
              flush=True)
        ref2 = ref0 + r.float()
        got2 = hip.gemm256_w(a, b, 2, r, None, 0, 0)
        print(f"M{m}: epi2 resid relerr={relerr(got2, ref2):.3e}", flush=True)
        # determinism
        outs = [hip.gemm256_w(a, b, 0, None, None, 0, 0) for _ in range(3)]
        print(f"M{m}: self-stable={all(torch.equal(outs[0], o) for o in outs[1:])}",
              flush=True)

    # ---- perf ----
    shapes = [("qkv", M, 6144, 4096), ("gateup", M, 28672, 4096),
              ("down", M, 4096, 14336)]
    for name, m, n, k in shapes:
        a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        flops = 2.0 * m * n * k
        variants = {
            "b16": lambda: ops.gemm256_nt(a, b),
            "w32": lambda: hip.gemm256_w(a, b, 0, None, None, 0, 0),
            "blaslt": lambda: torch.matmul(a, b.t()),
        }
        for fn in variants.values():
            for _ in range(3):
                fn()
        acc = {kk: [] for kk in variants}
        for _ in range(5):
            for kk, fn in variants.items():
                acc[kk].append(timed(fn))
        out = {kk: flops / min(v) / 1e12 for kk, v in acc.items()}
        print(f"{name:8s} M{m} N{n} K{k} "
              + " ".join(f"{kk}={tf:7.1f}TF" for kk, tf in out.items()), flush=True)


if __name__ == "__main__":
    main()
