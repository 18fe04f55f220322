"""Decode-attention microbench: per-call time of the two-pass chunked
kernel vs L, isolating the fixed overhead NEXT.md #3 describes.
GPU box:  python tests/attn_decode_perf_driver.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops


def timed(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    torch.manual_seed(3)
    dev = "cuda"
    B, Hq, Hkv, D, Smax = 1, 32, 8, 128, 8192
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.2
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.2
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=dev) * 0.2
    for L in (256, 512, 1024, 2048, 4096, 8192):
        L_dev = torch.tensor([L], dtype=torch.int32, device=dev)
        us_t = timed(lambda: ops.attn_decode_t(q, kc, vc, L_dev))
        us_h = timed(lambda: ops.attn_decode(q, kc, vc, L))
        mb = 2 * B * Hkv * L * D * 2 / 1e6
        print(f"L={L:5d}  L_dev={us_t:7.1f}us  host-L={us_h:7.1f}us  "
              f"kv={mb:6.1f}MB  eff={mb/1e3/(us_t*1e-6):6.0f}GB/s", flush=True)
    # chunk-vs-single routing A/B at larger batches (the G-sharing
    # two-pass form reads each KV cache once; single-pass reads it G x)
    import os
    for B in (8, 32):
        kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.2
        vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.2
        q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=dev) * 0.2
        for L in (512, 2048):
            L_dev = torch.tensor([L], dtype=torch.int32, device=dev)
            res = {}
            for mode in ("chunk", "single"):
                os.environ["BOBRA_DEC_ATTN"] = mode
                res[mode] = timed(lambda: ops.attn_decode_t(q, kc, vc, L_dev))
            os.environ.pop("BOBRA_DEC_ATTN", None)
            print(f"b{B} L={L:5d}: chunk={res['chunk']:6.1f}us "
                  f"single={res['single']:6.1f}us", flush=True)

    # b8 shape (bench serving batch) for reference
    B = 8
    kc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.2
    vc = torch.randn(B, Hkv, Smax, D, dtype=torch.bfloat16, device=dev) * 0.2
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device=dev) * 0.2
    for L in (1024, 4096):
        L_dev = torch.tensor([L], dtype=torch.int32, device=dev)
        us_t = timed(lambda: ops.attn_decode_t(q, kc, vc, L_dev))
        mb = 2 * B * Hkv * L * D * 2 / 1e6
        print(f"b8 L={L:5d}  L_dev={us_t:7.1f}us  kv={mb:6.1f}MB  "
              f"eff={mb/1e3/(us_t*1e-6):6.0f}GB/s", flush=True)


if __name__ == "__main__":
    main()
