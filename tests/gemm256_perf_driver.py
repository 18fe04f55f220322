"""GEMM-256 perf A/B vs hipBLASLt (torch.matmul) on the Llama-3-8B prefill
shapes.  Run on a GPU box:
    python tests/gemm256_perf_driver.py [M]
Prints per-shape ms + TF/s for (a) torch.matmul NT, (b) gemm256 plain,
(c) the fused-epilogue variant the model would use.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
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
    shapes = [
        ("qkv", M, 6144, 4096, "scale"),
        ("o", M, 4096, 4096, "resid"),
        ("gateup", M, 28672, 4096, "swiglu"),
        ("down", M, 4096, 14336, "resid"),
        ("lm_head", 4 if M > 4 else M, 128256, 4096, "plain"),
    ]
    for name, m, n, k, epi in shapes:
        a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        flops = 2.0 * m * n * k
        t_lib = bench(lambda: torch.matmul(a, b.t()))
        t_own = bench(lambda: ops.gemm256_nt(a, b))
        if epi == "scale":
            stat = ops.rowsumsq(a)
            t_epi = bench(lambda: ops.gemm256_nt(a, b, stat, 1.0 / k, 1e-5))
        elif epi == "swiglu":
            stat = ops.rowsumsq(a)
            t_epi = bench(lambda: ops.gemm256_swiglu(a, b, stat, 1.0 / k, 1e-5))
        elif epi == "resid":
            r = torch.randn(m, n, dtype=torch.bfloat16, device=dev)
            t_epi = bench(lambda: ops.gemm256_resid(a, b, r, True))
        else:
            t_epi = t_own
        # correctness spot check (plain)
        got = ops.gemm256_nt(a[: min(m, 512)], b)
        ref = torch.matmul(a[: min(m, 512)].float(), b.float().t())
        rel = (got.float() - ref).abs().max().item() / max(ref.abs().max().item(), 1.0)
        print(
            f"{name:8s} M={m:6d} N={n:6d} K={k:6d}  "
            f"lib {t_lib*1e3:7.3f} ms ({flops/t_lib/1e12:7.1f} TF)  "
            f"own {t_own*1e3:7.3f} ms ({flops/t_own/1e12:7.1f} TF)  "
            f"epi {t_epi*1e3:7.3f} ms ({flops/t_epi/1e12:7.1f} TF)  relerr {rel:.4f}",
            flush=True,
        )


if __name__ == "__main__":
    main()
