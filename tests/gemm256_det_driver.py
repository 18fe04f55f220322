"""Determinism probe for gemm256b disciplines: each variant run twice on
identical inputs must be bitwise-identical with itself (a mismatch = a
race); also quantifies disc0 vs disc1 cross-variant differences.
GPU box:  python tests/gemm256_det_driver.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops

hip = ops._try_load()
assert hip is not None, ops._load_error


def main():
    torch.manual_seed(11)
    dev = "cuda"
    for (m, n, k) in ((512, 6144, 4096), (8192, 6144, 4096), (8192, 4096, 14336)):
        a = (torch.randn(m, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        b = (torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.3).contiguous()
        ref = torch.matmul(a.float(), b.float().t())
        for name, fn in (("disc0", lambda: ops.gemm256_nt(a, b)),
                         ("disc1", lambda: hip.gemm256_nt_disc(a, b, 1)),
                         ("disc2", lambda: hip.gemm256_nt_disc(a, b, 2))):
            outs = [fn() for _ in range(4)]
            stable = all(torch.equal(outs[0], o) for o in outs[1:])
            err = (outs[0].float() - ref).abs()
            rel = (err.max() / ref.abs().max()).item()
            print(f"M{m} N{n} K{k} {name}: self-stable={stable} "
                  f"relerr_vs_fp32={rel:.3e}", flush=True)
        d0 = ops.gemm256_nt(a, b)
        d1 = hip.gemm256_nt_disc(a, b, 1)
        diff = (d0 != d1)
        nd = int(diff.sum())
        if nd:
            idx = diff.nonzero()[:5]
            mx = (d0.float() - d1.float()).abs().max().item()
            print(f"  cross diff: {nd}/{d0.numel()} elems, max |d0-d1|={mx:.3e}, "
                  f"first rows={idx[:, 0].tolist()} cols={idx[:, 1].tolist()}",
                  flush=True)
        else:
            print("  cross: identical", flush=True)


if __name__ == "__main__":
    main()
