"""Llama model numerics: the fused serving layout (norm gains folded into
weights, interleaved gate/up, row-scale/SwiGLU/residual epilogues) must
match the classic unfused transformer math.  CPU (fp32 reference ops);
the GPU equivalents live in tests/test_ops_gpu.py / test_gpu_e2e.py."""
import math

import pytest
import torch

from bobrapet_amd.models.llama import (
    CONFIGS,
    LlamaModel,
    fold_gain,
    interleave_gate_up,
)
from bobrapet_amd import ops


def _ref_prefill(model, raw, ids):
    """Classic unfused math from the RAW (unfolded) weights."""
    cfg = model.cfg
    B, S = ids.shape
    eps = cfg.rms_eps

    def rms(x, g):
        ms = x.pow(2).mean(-1, keepdim=True)
        return x * torch.rsqrt(ms + eps) * g

    x = model.embed[ids].float()
    pos = torch.arange(S)
    cos_t, sin_t = ops.rope_tables(pos, cfg.head_dim, cfg.rope_theta)

    for li in range(cfg.num_layers):
        w = raw[li]
        h = rms(x, w["ln_attn"])
        qkv = h @ w["w_qkv"].t()
        nq = cfg.num_heads * cfg.head_dim
        nkv = cfg.num_kv_heads * cfg.head_dim
        q = qkv[..., :nq].reshape(B * S, cfg.num_heads, cfg.head_dim)
        k = qkv[..., nq : nq + nkv].reshape(B * S, cfg.num_kv_heads, cfg.head_dim)
        v = qkv[..., nq + nkv :].reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
        q = ops.rope_ref(q, cos_t.repeat(B, 1), sin_t.repeat(B, 1)).view(B, S, -1, cfg.head_dim)
        k = ops.rope_ref(k, cos_t.repeat(B, 1), sin_t.repeat(B, 1)).view(B, S, -1, cfg.head_dim)
        attn = ops.attn_ref(
            q.float().permute(0, 2, 1, 3),
            k.float().permute(0, 2, 1, 3),
            v.float().permute(0, 2, 1, 3),
            1.0 / math.sqrt(cfg.head_dim),
            True,
        ).permute(0, 2, 1, 3)
        x = x + attn.reshape(B, S, -1).float() @ w["w_o"].t()
        h = rms(x, w["ln_mlp"])
        gu = h @ w["w_gate_up"].t()
        g, u = gu.chunk(2, dim=-1)
        x = x + (g * torch.sigmoid(g) * u) @ w["w_down"].t()
    h = rms(x, raw["ln_final"])
    return h[:, -1] @ raw["lm_head"].t()


@pytest.fixture
def tiny_folded():
    torch.manual_seed(5)
    m = LlamaModel("llama-tiny", device="cpu")
    cfg = m.cfg
    H, I = cfg.hidden_size, cfg.intermediate_size
    raw = {}
    for li, lw in enumerate(m.layers):
        w = {
            "ln_attn": torch.rand(H) * 0.5 + 0.75,   # NON-unit gains
            "ln_mlp": torch.rand(H) * 0.5 + 0.75,
            "w_qkv": torch.randn(cfg.qkv_out, H) * 0.05,
            "w_o": torch.randn(H, cfg.num_heads * cfg.head_dim) * 0.05,
            "w_gate_up": torch.randn(2 * I, H) * 0.05,
            "w_down": torch.randn(H, I) * 0.05,
        }
        raw[li] = w
        lw.ln_attn = w["ln_attn"].bfloat16()
        lw.ln_mlp = w["ln_mlp"].bfloat16()
        lw.w_qkv = fold_gain(w["w_qkv"].bfloat16(), lw.ln_attn)
        lw.w_o = w["w_o"].bfloat16()
        lw.w_gate_up = fold_gain(
            interleave_gate_up(w["w_gate_up"].bfloat16()), lw.ln_mlp
        )
        lw.w_down = w["w_down"].bfloat16()
    raw["ln_final"] = torch.rand(H) * 0.5 + 0.75
    raw["lm_head"] = torch.randn(cfg.vocab_size, H) * 0.05
    m.ln_final = raw["ln_final"].bfloat16()
    m.lm_head = fold_gain(raw["lm_head"].bfloat16(), m.ln_final)
    # raw refs in fp32 reflecting the bf16 storage the model actually uses
    for li in raw:
        if isinstance(li, int):
            for k in raw[li]:
                raw[li][k] = raw[li][k].bfloat16().float()
    raw["ln_final"] = raw["ln_final"].bfloat16().float()
    raw["lm_head"] = raw["lm_head"].bfloat16().float()
    return m, raw


def test_fused_prefill_matches_unfused_reference(tiny_folded):
    m, raw = tiny_folded
    ids = torch.randint(0, m.cfg.vocab_size, (2, 32))
    got = m.prefill(ids).float()
    ref = _ref_prefill(m, raw, ids)
    scale = ref.abs().max().item()
    assert (got - ref).abs().max().item() < 0.05 * max(scale, 1.0)


def test_prefill_matches_decode(tiny_folded):
    m, _ = tiny_folded
    ids = torch.randint(0, m.cfg.vocab_size, (1, 33))
    full = m.prefill(ids).float()
    m2, _ = tiny_folded.__class__ and (m, None)  # same model, fresh cache
    _ = m.prefill(ids[:, :32], fill_cache=True)
    dec = m.decode_step(ids[:, 32]).float()
    scale = full.abs().max().item()
    assert (dec - full).abs().max().item() < 0.08 * max(scale, 1.0)


def test_interleave_roundtrip():
    w = torch.arange(24, dtype=torch.float32).reshape(6, 4)
    inter = interleave_gate_up(w)
    assert torch.equal(inter[0], w[0])   # gate_0
    assert torch.equal(inter[1], w[3])   # up_0
    assert torch.equal(inter[4], w[2])   # gate_2
    assert torch.equal(inter[5], w[5])   # up_2


def test_all_configs_satisfy_gemm_alignment():
    for name, cfg in CONFIGS.items():
        for n, k in (
            (cfg.qkv_out, cfg.hidden_size),
            (cfg.hidden_size, cfg.num_heads * cfg.head_dim),
            (2 * cfg.intermediate_size, cfg.hidden_size),
            (cfg.hidden_size, cfg.intermediate_size),
            (cfg.vocab_size, cfg.hidden_size),
        ):
            assert n % 256 == 0 and k % 32 == 0, (name, n, k)
