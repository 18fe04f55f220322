"""Serving metrics snapshot: TTFT (prefill b1) and decode throughput at
several batch sizes, 8B bf16.  GPU box: python tests/serving_metrics_driver.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bobrapet_amd.models.llama import LlamaModel


def main():
    m = LlamaModel("llama-3-8b", device="cuda")
    # TTFT: prefill b1 at a few context lengths
    for S in (512, 2048, 8192):
        ids = torch.randint(0, m.cfg.vocab_size, (1, S), device="cuda")
        for _ in range(2):
            m.prefill(ids)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            m.prefill(ids)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / 5 * 1e3
        print(f"prefill b1 s{S}: {ms:7.1f} ms  ({S/ms*1e3:7.0f} tok/s)", flush=True)
    # decode throughput at b1/b8/b32
    for B in (1, 8, 32):
        ids = torch.randint(0, m.cfg.vocab_size, (B, 512), device="cuda")
        m.prefill(ids, fill_cache=True)
        nxt = torch.randint(0, m.cfg.vocab_size, (B,), device="cuda")
        for _ in range(10):
            nxt = m.decode_step_graphed(nxt).argmax(-1)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(100):
            nxt = m.decode_step_graphed(nxt).argmax(-1)
        torch.cuda.synchronize()
        ms = (time.perf_counter() - t0) / 100 * 1e3
        print(f"decode b{B}: {ms:6.2f} ms/step = {B/ms*1e3:7.0f} tok/s", flush=True)


if __name__ == "__main__":
    main()
