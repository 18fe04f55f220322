"""Sweep the gemv2 persistent-grid cap on the decode shapes.
GPU box:  python tests/gemv_cap_driver.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops


def timed(fn, iters=300, warmup=30):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    torch.manual_seed(2)
    dev = "cuda"
    K = 4096
    x = torch.randn(1, K, dtype=torch.bfloat16, device=dev) * 0.3
    shapes = {
        "qkv6144": (ops.gemv_norm, torch.randn(6144, K, dtype=torch.bfloat16, device=dev) * 0.05, 50.3),
        "o4096": (None, torch.randn(4096, K, dtype=torch.bfloat16, device=dev) * 0.05, 33.6),
        "gateup": (None, torch.randn(28672, K, dtype=torch.bfloat16, device=dev) * 0.05, 235.0),
        "down": (None, torch.randn(4096, 14336, dtype=torch.bfloat16, device=dev) * 0.05, 117.4),
    }
    xd = torch.randn(1, 14336, dtype=torch.bfloat16, device=dev) * 0.3
    r = torch.randn(1, 4096, dtype=torch.bfloat16, device=dev)
    for cap in ("2048", "1024", "768", "512", "384"):
        os.environ["BOBRA_GEMV_CAP"] = cap
        t_qkv = timed(lambda: ops.gemv_norm(x, shapes["qkv6144"][1], 1.0 / K, 1e-5))
        t_o = timed(lambda: ops.gemv_resid(x, shapes["o4096"][1], r))
        t_gu = timed(lambda: ops.gemv_swiglu_norm(x, shapes["gateup"][1], 1.0 / K, 1e-5))
        t_dn = timed(lambda: ops.gemv_resid(xd, shapes["down"][1], r))
        tot = t_qkv + t_o + t_gu + t_dn
        print(f"cap={cap:>5}: qkv={t_qkv:5.1f}us ({50.3/1e3/(t_qkv*1e-6):4.0f}GB/s) "
              f"o={t_o:5.1f} gu={t_gu:5.1f} ({235/1e3/(t_gu*1e-6):4.0f}GB/s) "
              f"dn={t_dn:5.1f} layer_total={tot:6.1f}us", flush=True)
    # numerics guard at the best-looking cap
    os.environ["BOBRA_GEMV_CAP"] = "512"
    got = ops.gemv_norm(x, shapes["qkv6144"][1], 1.0 / K, 1e-5)
    xf = x.float()
    s = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5)
    ref = (xf * s) @ shapes["qkv6144"][1].float().t()
    rel = ((got.float() - ref).abs().max() / ref.abs().max()).item()
    print(f"cap=512 numerics relerr={rel:.3e}", flush=True)


if __name__ == "__main__":
    main()
