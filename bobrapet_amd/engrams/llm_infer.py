"""llm-infer engram: transformer inference on MI355X.

The flagship built-in engram (BASELINE.json config #2: batch Story with one
llm-infer engram, Llama-3-8B bf16, 1 MI355X).  Weights are random-init
(no network for checkpoints) and stay resident in HBM across steps via the
process-wide model cache; the step launches on its placed (device, stream)
worker slot.

config (engram `with`): {model: llama-3-8b | llama-tiny, seed}
input (step `with`):
  {batch: int, seqLen: int, newTokens: int (default 0 = prefill only),
   promptIds: optional [[int]] explicit prompts; otherwise synthetic}
output: {model, batch, seqLen, newTokens, tokens, latencyMs, logits: $storageRef}
"""
from __future__ import annotations

import time

import torch

from .base import Engram, EngramContext, EngramFailure, EngramResult
from .registry import register_class


@register_class
class LlmInferEngram(Engram):
    name = "llm-infer"
    wants_gpu = True

    def run(self, ctx: EngramContext) -> EngramResult:
        from ..models.llama import CONFIGS, get_model

        cfg = dict(ctx.config or {})
        inp = dict(ctx.input or {}) if isinstance(ctx.input, dict) else {}
        model_name = inp.get("model") or cfg.get("model") or "llama-3-8b"
        if model_name not in CONFIGS:
            raise EngramFailure(f"unknown model {model_name!r}", exit_code=2)
        batch = int(inp.get("batch", cfg.get("batch", 1)))
        seq_len = int(inp.get("seqLen", cfg.get("seqLen", 128)))
        new_tokens = int(inp.get("newTokens", cfg.get("newTokens", 0)))

        device = f"cuda:{ctx.device}" if ctx.device is not None else "cpu"
        if device == "cpu" and torch.cuda.is_available():
            raise EngramFailure("llm-infer placed on CPU with GPUs present", exit_code=2)
        model = get_model(model_name, device=device)

        prompt = inp.get("promptIds")
        if prompt is not None:
            ids = torch.tensor(prompt, dtype=torch.long, device=model.device)
            batch, seq_len = ids.shape
        else:
            gen = torch.Generator(device="cpu").manual_seed(
                int(inp.get("seed", cfg.get("seed", 0)))
            )
            ids = torch.randint(
                0, model.cfg.vocab_size, (batch, seq_len), generator=gen
            ).to(model.device)

        # sequenceParallel: true shards the sequence across the live
        # distributed ranks (SPMD story) — ring attention per layer; each
        # rank's `ids` must be its rank-major shard of the global sequence
        seq_parallel = bool(inp.get("sequenceParallel", cfg.get("sequenceParallel", False)))
        seq_shard = None
        if seq_parallel:
            from ..parallel import group

            if group.world_size() > 1:
                seq_shard = (group.rank(), group.world_size())

        t0 = time.monotonic()
        if new_tokens > 0:
            tokens = model.generate(ids, new_tokens)
            logits = None
        else:
            logits = model.prefill(ids, seq_shard=seq_shard)
            tokens = logits.argmax(dim=-1, keepdim=True)
        if model.device.type == "cuda":
            torch.cuda.synchronize(model.device)
        latency_ms = (time.monotonic() - t0) * 1000.0

        out = {
            "model": model_name,
            "batch": batch,
            "seqLen": seq_len,
            "newTokens": new_tokens,
            "tokens": tokens[:, :32].tolist(),  # inline sample
            "latencyMs": latency_ms,
            "tokensProcessed": batch * (seq_len + new_tokens),
            "sequenceParallel": seq_shard is not None,
        }
        if logits is not None and ctx.storage is not None:
            # keep the full logits tensor resident (HBM payload indirection)
            out["logits"] = ctx.storage.offload_tensor(logits)
        return EngramResult(output=out)
