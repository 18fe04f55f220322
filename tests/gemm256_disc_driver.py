"""A/B probe: shipped gemm256b sync discipline (DISC=0, barriers at q0/q2)
vs the guide-template per-phase barrier-pair discipline (DISC=1).
Within-process interleaved rounds (guide rule 24).  GPU box:
    python tests/gemm256_disc_driver.py [M]
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops

hip = ops._try_load()
assert hip is not None, ops._load_error


def timed(fn, iters):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    M = int(sys.argv[1]) if len(sys.argv) > 1 else 8192
    torch.manual_seed(7)
    dev = "cuda"
    shapes = [("qkv", M, 6144, 4096), ("gateup", M, 28672, 4096),
              ("down", M, 4096, 14336)]
    for name, m, n, k in shapes:
        a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        flops = 2.0 * m * n * k
        # numerics: all disciplines must agree exactly (same math order)
        d0 = ops.gemm256_nt(a[:512], b)
        same = all(torch.equal(d0, hip.gemm256_nt_disc(a[:512], b, d))
                   for d in (1, 2, 3))
        variants = {
            "disc0": lambda: ops.gemm256_nt(a, b),
            "disc1": lambda: hip.gemm256_nt_disc(a, b, 1),
            "disc2": lambda: hip.gemm256_nt_disc(a, b, 2),
            "disc3": lambda: hip.gemm256_nt_disc(a, b, 3),
            "blaslt": lambda: torch.matmul(a, b.t()),
        }
        for fn in variants.values():  # warmup
            for _ in range(3):
                fn()
        acc = {kk: [] for kk in variants}
        for _ in range(5):  # interleaved rounds
            for kk, fn in variants.items():
                acc[kk].append(timed(fn, 10))
        out = {kk: flops / min(v) / 1e12 for kk, v in acc.items()}
        print(f"{name:8s} M{m} N{n} K{k} exact_match={same} "
              + " ".join(f"{kk}={tf:7.1f}TF" for kk, tf in out.items()),
              flush=True)


if __name__ == "__main__":
    main()
