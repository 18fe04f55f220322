"""A/B: lockstep attn_prefill vs the pipelined variant (softmax+PV of
tile t-1 under tile t's QK MFMAs).  Numerics vs the fp32 reference, then
interleaved perf rounds at the Llama-3-8B prefill shape.
GPU box:  python tests/attn_pipe_driver.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd import ops

hip = ops._try_load()
assert hip is not None, ops._load_error


def ref_attn(q, k, v, causal):
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3).repeat_interleave(Hq // Hkv, 1)
    vf = v.float().permute(0, 2, 1, 3).repeat_interleave(Hq // Hkv, 1)
    s = torch.matmul(qf, kf.transpose(-1, -2)) / D ** 0.5
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
        s = s.masked_fill(mask, float("-inf"))
    return torch.matmul(torch.softmax(s, -1), vf).permute(0, 2, 1, 3)


def timed(fn, iters=20):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    torch.manual_seed(5)
    dev = "cuda"
    scale = 1.0 / 128 ** 0.5
    # numerics at a mixed/edge shape first
    for (B, S, Hq, Hkv, causal) in ((2, 333, 8, 2, True), (1, 512, 32, 8, True),
                                    (1, 512, 32, 8, False)):
        q = torch.randn(B, S, Hq, 128, dtype=torch.bfloat16, device=dev) * 0.5
        k = torch.randn(B, S, Hkv, 128, dtype=torch.bfloat16, device=dev) * 0.5
        v = torch.randn(B, S, Hkv, 128, dtype=torch.bfloat16, device=dev) * 0.5
        ref = ref_attn(q, k, v, causal)
        for name, fn in (("lock", lambda: ops.attn_prefill(q, k, v, causal=causal)),
                         ("pipe", lambda: hip.attn_prefill_pipe(q, k, v, scale, causal))):
            got = fn().float()
            err = (got - ref).abs().max().item()
            denom = ref.abs().max().item()
            print(f"B{B} S{S} Hq{Hq} causal={causal} {name}: "
                  f"relerr={err / denom:.3e}", flush=True)
    # perf: llama-3-8b prefill shape
    B, S, Hq, Hkv = 4, 2048, 32, 8
    q = torch.randn(B, S, Hq, 128, dtype=torch.bfloat16, device=dev) * 0.5
    k = torch.randn(B, S, Hkv, 128, dtype=torch.bfloat16, device=dev) * 0.5
    v = torch.randn(B, S, Hkv, 128, dtype=torch.bfloat16, device=dev) * 0.5
    for causal in (True, False):
        fl = 4.0 * B * Hq * S * S * 128 * (0.5 if causal else 1.0)
        variants = {
            "lock": lambda: ops.attn_prefill(q, k, v, causal=causal),
            "pipe": lambda: hip.attn_prefill_pipe(q, k, v, scale, causal),
        }
        for fn in variants.values():
            for _ in range(5):
                fn()
        acc = {kk: [] for kk in variants}
        for _ in range(5):
            for kk, fn in variants.items():
                acc[kk].append(timed(fn))
        msg = " ".join(
            f"{kk}={min(v1) * 1e3:6.3f}ms({fl / min(v1) / 1e12:5.0f}TF)"
            for kk, v1 in acc.items())
        print(f"b{B} s{S} causal={causal}: {msg}", flush=True)


if __name__ == "__main__":
    main()
