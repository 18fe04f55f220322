"""Run one gemm256 shape in a loop for rocprofv3 PMC collection.
Usage: python tests/gemm256_pmc_driver.py [shape] [iters]
shape in {qkv, o, gateup, down}.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops

SHAPES = {
    "qkv": (8192, 6144, 4096),
    "o": (8192, 4096, 4096),
    "gateup": (8192, 28672, 4096),
    "down": (8192, 4096, 14336),
}


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "gateup"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 10
    variant = sys.argv[3] if len(sys.argv) > 3 else "b"
    m, n, k = SHAPES[name]
    torch.manual_seed(3)
    a = (torch.randn(m, k, dtype=torch.bfloat16, device="cuda") * 0.3).contiguous()
    b = (torch.randn(n, k, dtype=torch.bfloat16, device="cuda") * 0.3).contiguous()
    if variant == "w":
        hip = ops._try_load()
        for _ in range(iters):
            hip.gemm256_w(a, b, 0, None, None, 0, 0)
    else:
        for _ in range(iters):
            ops.gemm256_nt(a, b)
    torch.cuda.synchronize()
    print(f"done {name} x{iters} variant={variant}")


if __name__ == "__main__":
    main()
