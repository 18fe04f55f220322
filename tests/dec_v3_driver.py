"""Micro-bench: decode attention v3 (two-pass lane-per-row) vs v1 single-pass,
then end-to-end decode tok/s at b1/b32. Run on a GPU box."""
import os
import sys
import time

import torch

sys.path.insert(0, ".")
from bobrapet_amd import ops  # noqa: E402


def time_us(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def micro():
    torch.manual_seed(0)
    ext = ops._require_ext()
    for B, L in [(1, 512), (1, 2048), (8, 1024), (32, 2048)]:
        Hq, Hkv, D, Smax = 32, 8, 128, 4096
        q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
        kc = torch.randn(B, Hkv, Smax, D, device="cuda", dtype=torch.bfloat16)
        vc = torch.randn_like(kc)
        ref = ops.attn_decode_ref(q.float(), kc.float(), vc.float(), L)
        out = ext.attn_decode(q, kc, vc, L, D ** -0.5)
        err = (out.float() - ref).abs().max().item()
        t3 = time_us(lambda: ext.attn_decode(q, kc, vc, L, D ** -0.5))
        os.environ["BOBRA_DEC_V1"] = "1"
        out1 = ext.attn_decode(q, kc, vc, L, D ** -0.5)
        err1 = (out1.float() - ref).abs().max().item()
        t1 = time_us(lambda: ext.attn_decode(q, kc, vc, L, D ** -0.5))
        del os.environ["BOBRA_DEC_V1"]
        print(
            f"B={B:3d} L={L:5d}  v3={t3:7.1f}us (err {err:.4f})  "
            f"v1={t1:7.1f}us (err {err1:.4f})  speedup {t1 / t3:.2f}x"
        )
        assert err < 0.05, "v3 numerics"


def e2e():
    from bobrapet_amd.models.llama import LlamaModel

    for B in (1, 32):
        m = LlamaModel("llama-3-8b", device="cuda")
        ids = torch.randint(0, m.cfg.vocab_size, (B, 512), device="cuda")
        m.prefill(ids, fill_cache=True)
        nxt = torch.randint(0, m.cfg.vocab_size, (B,), device="cuda")
        for _ in range(3):
            nxt = m.decode_step_graphed(nxt).argmax(-1)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        n = 50
        for _ in range(n):
            nxt = m.decode_step_graphed(nxt).argmax(-1)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / n * 1e3
        print(f"decode b{B}: {ms:.2f} ms/step  {B / ms * 1e3:.0f} tok/s")
        del m
        torch.cuda.empty_cache()


if __name__ == "__main__":
    micro()
    e2e()
    print("OK")
