"""bobrapet_amd.ops — the hand-written CDNA4 kernel library.

Dispatch rules:
  - GPU tensors → the in-tree `_hipops` extension (hipcc-built, gfx950).
    If the extension is missing on a GPU machine, ops FAIL LOUDLY — there
    is no silent eager fallback on the GPU path.
  - CPU tensors → plain fp32 torch reference implementations (the same
    code the numerics tests compare the HIP kernels against).
"""
from __future__ import annotations

import math
import os
import typing as _t

import torch

_hipops = None
_load_error: _t.Optional[str] = None


def _try_load():
    global _hipops, _load_error
    if _hipops is not None:
        return _hipops
    try:
        from bobrapet_amd import _hipops as ext  # in-tree .so

        _hipops = ext
    except ImportError as exc:
        _load_error = str(exc)
    return _hipops


def hip_available() -> bool:
    return _try_load() is not None


def _require_ext():
    ext = _try_load()
    if ext is None:
        raise RuntimeError(
            "bobrapet_amd._hipops is not built but a GPU tensor was passed. "
            "Build it with `python -m bobrapet_amd.csrc.build` "
            f"(import error: {_load_error})"
        )
    return ext


# ---------------------------------------------------------------------------
# reference implementations (fp32 torch, CPU or GPU) — used on CPU and as
# the comparison baseline in tests/test_ops_gpu.py
# ---------------------------------------------------------------------------


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    ms = xf.pow(2).mean(dim=-1, keepdim=True)
    return (xf * torch.rsqrt(ms + eps) * w.float()).to(x.dtype)


def fused_add_rmsnorm_ref(x, residual, w, eps=1e-5):
    res = (residual.float() + x.float()).to(x.dtype)
    return rmsnorm_ref(res, w, eps), res


def silu_mul_ref(gate, up):
    gf = gate.float()
    return (gf * torch.sigmoid(gf) * up.float()).to(gate.dtype)


def rope_tables(positions: torch.Tensor, dim: int, theta: float = 500000.0):
    """cos/sin tables [T, dim/2] f32 (host-precomputed, guide App. B)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, dim, 2, dtype=torch.float32, device=positions.device) / dim)
    )
    ang = positions.float()[:, None] * inv_freq[None, :]
    return torch.cos(ang).contiguous(), torch.sin(ang).contiguous()


def rope_ref(x: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor) -> torch.Tensor:
    """NeoX half-rotation; x: [T, H, D], tables [T, D/2]."""
    xf = x.float()
    half = x.shape[-1] // 2
    a, b = xf[..., :half], xf[..., half:]
    c = cos_t[:, None, :]
    s = sin_t[:, None, :]
    return torch.cat([a * c - b * s, b * c + a * s], dim=-1).to(x.dtype)


def embed_pool_ref(table: torch.Tensor, ids: torch.Tensor) -> torch.Tensor:
    g = table.float()[ids.long()]  # [B,S,H]
    pooled = g.mean(dim=1)
    return torch.nn.functional.normalize(pooled, dim=-1, eps=1e-6).to(table.dtype)


def attn_ref(q, k, v, scale: _t.Optional[float] = None, causal: bool = True):
    """[B,H,S,D] fp32 reference attention with GQA expansion."""
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    qf, kf, vf = q.float(), k.float(), v.float()
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), diagonal=1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, vf).to(q.dtype)


def attn_decode_ref(q, kc, vc, L: int, scale: _t.Optional[float] = None):
    """q [B,Hq,D]; caches [B,Hkv,Smax,D] → [B,Hq,D]."""
    B, Hq, D = q.shape
    Hkv = kc.shape[1]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    k = kc[:, :, :L].float()
    v = vc[:, :, :L].float()
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    s = torch.einsum("bhd,bhld->bhl", q.float(), k) * scale
    p = torch.softmax(s, dim=-1)
    return torch.einsum("bhl,bhld->bhd", p, v).to(q.dtype)


# ---------------------------------------------------------------------------
# dispatching public ops
# ---------------------------------------------------------------------------


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5) -> torch.Tensor:
    if x.is_cuda:
        return _require_ext().rmsnorm(x.contiguous(), w.contiguous(), eps)
    return rmsnorm_ref(x, w, eps)


def fused_add_rmsnorm(x, residual, w, eps: float = 1e-5):
    """residual ← residual + x (in place on GPU); returns (normed, residual)."""
    if x.is_cuda:
        out = _require_ext().fused_add_rmsnorm(
            x.contiguous(), residual, w.contiguous(), eps
        )
        return out, residual
    return fused_add_rmsnorm_ref(x, residual, w, eps)


def silu_mul(gate, up):
    if gate.is_cuda:
        return _require_ext().silu_mul(gate.contiguous(), up.contiguous())
    return silu_mul_ref(gate, up)


def rope_inplace(q, k, cos_t, sin_t):
    """q [T,Hq,D], k [T,Hk,D] rotated in place (GPU) / returned (CPU)."""
    if q.is_cuda:
        _require_ext().rope_inplace(q, k, cos_t, sin_t)
        return q, k
    return rope_ref(q, cos_t, sin_t), rope_ref(k, cos_t, sin_t)


def embed_pool(table, ids):
    if table.is_cuda:
        return _require_ext().embed_pool(table.contiguous(), ids.to(torch.int32).contiguous())
    return embed_pool_ref(table, ids)


def add_bf16(a, b):
    if a.is_cuda:
        return _require_ext().add_bf16(a.contiguous(), b.contiguous())
    return (a.float() + b.float()).to(a.dtype)


def attn_prefill_stats_ref(q, k, v, scale, causal):
    """fp32 reference of attn_prefill_stats: BSHD in, (out, stats) out.
    stats[..., 0] = m, [..., 1] = l in the KERNEL's exp2 domain
    (scores * scale * log2(e)) so cross-block merges are implementation-
    uniform on CPU and GPU."""
    import torch

    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    g = Hq // Hkv
    qt = q.float().permute(0, 2, 1, 3)  # [B,Hq,S,D]
    kt = k.float().permute(0, 2, 1, 3)
    vt = v.float().permute(0, 2, 1, 3)
    if g > 1:
        kt = kt.repeat_interleave(g, dim=1)
        vt = vt.repeat_interleave(g, dim=1)
    s2 = torch.matmul(qt, kt.transpose(-1, -2)) * (scale * 1.4426950408889634)
    if causal:
        mask = torch.ones(S, k.shape[1], dtype=torch.bool, device=q.device).tril_()
        s2 = s2.masked_fill(~mask, float("-inf"))
    m = s2.amax(dim=-1)  # [B,Hq,S]
    p = torch.exp2(s2 - m[..., None])
    l = p.sum(dim=-1)
    out = torch.matmul(p, vt) / l[..., None]
    stats = torch.stack([m, l], dim=-1)  # [B,Hq,S,2]
    return out.permute(0, 2, 1, 3).to(q.dtype).contiguous(), stats


def attn_prefill_stats(q, k, v, scale: _t.Optional[float] = None, causal: bool = True):
    """attn_prefill + per-row (m, l) softmax stats for cross-block merging
    (ring attention over xGMI).  BSHD; stats [B,Hq,S,2] f32 exp2-domain."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        if not (v.stride(-1) == 1 and v.stride(2) == v.shape[-1]):
            v = v.contiguous()
        out, stats = _require_ext().attn_prefill_stats(
            q.contiguous(), k.contiguous(), v, scale, causal
        )
        return out, stats
    return attn_prefill_stats_ref(q, k, v, scale, causal)


def attn_prefill(q, k, v, scale: _t.Optional[float] = None, causal: bool = True):
    """BSHD layout: q [B,S,Hq,D], k/v [B,S,Hkv,D] → out [B,S,Hq,D]."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        if not (v.stride(-1) == 1 and v.stride(2) == v.shape[-1]):
            v = v.contiguous()
        return _require_ext().attn_prefill(q.contiguous(), k.contiguous(), v, scale, causal)
    out = attn_ref(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2), scale, causal
    )
    return out.transpose(1, 2).contiguous()


def rope_qkv_split(qkv, B, S, Hq, Hkv, D, cos_t, sin_t):
    """Fused split+rope: qkv [B,S,(Hq+2*Hkv)*D] → rotated contiguous
    q [B,S,Hq,D], k [B,S,Hkv,D] and v as a STRIDED VIEW into qkv (the
    attention kernel reads strided v — no copies)."""
    rowlen = qkv.shape[-1]
    nq, nkv = Hq * D, Hkv * D
    if qkv.is_cuda:
        flat = qkv.reshape(B * S, rowlen)
        q, k = _require_ext().rope_qkv(flat, Hq, Hkv, D, cos_t, sin_t)
        v = qkv[..., nq + nkv :].view(B, S, Hkv, D)
        return q.view(B, S, Hq, D), k.view(B, S, Hkv, D), v
    q = qkv[..., :nq].reshape(B * S, Hq, D)
    k = qkv[..., nq : nq + nkv].reshape(B * S, Hkv, D)
    v = qkv[..., nq + nkv :].reshape(B, S, Hkv, D)
    q = rope_ref(q, cos_t, sin_t).view(B, S, Hq, D)
    k = rope_ref(k, cos_t, sin_t).view(B, S, Hkv, D)
    return q, k, v


def gemmsk_nt(a, b, stat, mul: float, eps: float):
    """Skinny-M (<=32) GEMM with the rowscale epilogue — the decode
    batch-9..32 projection path (csrc/hip/gemmsk.hip)."""
    if a.is_cuda:
        (c,) = _require_ext().gemmsk(a, b, 0, None, stat, mul, eps)
        return c
    s = torch.rsqrt(stat.float() * mul + eps)
    return ((a.float() @ b.float().t()) * s[:, None]).to(a.dtype)


def gemmsk_swiglu(a, b_interleaved, stat, mul: float, eps: float):
    if a.is_cuda:
        (c,) = _require_ext().gemmsk(a, b_interleaved, 1, None, stat, mul, eps)
        return c
    s = torch.rsqrt(stat.float() * mul + eps)
    c = (a.float() @ b_interleaved.float().t()) * s[:, None]
    g, u = c[:, 0::2], c[:, 1::2]
    return (torch.nn.functional.silu(g) * u).to(a.dtype)


def gemmsk_resid(a, b, resid):
    """out = a @ b.T + resid; also returns the row sumsq of out (the
    next projection's norm statistic)."""
    if a.is_cuda:
        c, stat = _require_ext().gemmsk(a, b, 2, resid, None, 0.0, 0.0)
        return c, stat
    c = (a.float() @ b.float().t() + resid.float())
    return c.to(a.dtype), c.pow(2).sum(-1)


def rope_k_from_qkv(qkv_flat, Hq, Hkv, D, cos_t, sin_t):
    """Rope ONLY the K heads out of fused qkv rows [T, rowlen] → k
    [T, Hkv, D] contiguous.  Pairs with attn_prefill_qrope (which ropes Q
    on load): Q never round-trips HBM through a rope kernel."""
    if qkv_flat.is_cuda:
        (k,) = _require_ext().rope_k_only(qkv_flat.contiguous(), Hq, Hkv, D,
                                          cos_t, sin_t)
        return k
    nq = Hq * D
    k = qkv_flat[..., nq : nq + Hkv * D].reshape(-1, Hkv, D)
    return rope_ref(k, cos_t, sin_t)


def attn_prefill_qrope(q_view, k, v, inv_freq, pos0, scale, causal=True):
    """Prefill attention with rope applied to Q ON LOAD from a strided
    view into the fused qkv projection (csrc/hip/attention.hip
    launch_attn_prefill_qrope).  k must already be roped
    (rope_k_from_qkv); inv_freq is the f32 [D/2] frequency vector and
    pos0 the sequence-shard offset."""
    if q_view.is_cuda:
        return _require_ext().attn_prefill_qrope(q_view, k, v, inv_freq,
                                                 int(pos0), float(scale),
                                                 bool(causal))
    B, S, Hq, D = q_view.shape
    pos = torch.arange(pos0, pos0 + S, dtype=torch.float32)
    ang = pos[:, None] * inv_freq.float()[None, :]
    qr = rope_ref(q_view.reshape(B * S, Hq, D), ang.cos().repeat(B, 1),
                  ang.sin().repeat(B, 1)).view(B, S, Hq, D)
    return attn_ref(qr.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                    scale, causal).transpose(1, 2)


def silu_mul_fused(gate_up: torch.Tensor) -> torch.Tensor:
    """silu(g)*u from the fused [.., 2I] gate_up projection (no slicing
    copies on GPU)."""
    if gate_up.is_cuda:
        return _require_ext().silu_mul_strided(gate_up.contiguous())
    gate, up = gate_up.chunk(2, dim=-1)
    return silu_mul_ref(gate, up)


def gemv_nt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """C = a @ b.T for a [M<=8, K]: the weight-streaming decode GEMV."""
    if a.is_cuda:
        return _require_ext().gemv_nt(a.contiguous(), b.contiguous())
    return (a.float() @ b.float().t()).to(a.dtype)


def gemm_nt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """C = a @ b.T with b stored [N, K] (nn.Linear weight layout).

    GPU: the hand-written MFMA kernel (csrc/hip/gemm.hip) for aligned
    shapes, else hipBLASLt via torch.matmul; CPU: fp32 reference."""
    if a.is_cuda:
        M, K = a.shape
        N = b.shape[0]
        if M % 128 == 0 and N % 128 == 0 and K % 64 == 0:
            return _require_ext().gemm_nt(a.contiguous(), b.contiguous())
        return torch.matmul(a, b.t())
    return (a.float() @ b.float().t()).to(a.dtype)


def _rowscale(stat: torch.Tensor, stat_mul: float, stat_eps: float) -> torch.Tensor:
    return torch.rsqrt(stat.float() * stat_mul + stat_eps)


def gemm256_nt(
    a: torch.Tensor,
    b: torch.Tensor,
    stat: _t.Optional[torch.Tensor] = None,
    stat_mul: float = 0.0,
    stat_eps: float = 0.0,
) -> torch.Tensor:
    """C = a @ b.T (b stored [N,K]) on the 256-tile MFMA GEMM; optional
    fused row-scale epilogue C[m] *= rsqrt(stat[m]*mul + eps) — the RMSNorm
    entry with the per-channel gain folded into b (csrc/hip/gemm256.hip)."""
    if a.is_cuda:
        return _require_ext().gemm256_nt(a.contiguous(), b.contiguous(), stat, stat_mul, stat_eps)
    c = a.float() @ b.float().t()
    if stat is not None:
        c = c * _rowscale(stat, stat_mul, stat_eps)[:, None]
    return c.to(a.dtype)


def gemm256_swiglu(
    a: torch.Tensor,
    b_interleaved: torch.Tensor,
    stat: _t.Optional[torch.Tensor] = None,
    stat_mul: float = 0.0,
    stat_eps: float = 0.0,
) -> torch.Tensor:
    """Fused gate/up projection + SwiGLU: b rows interleaved
    (gate_0, up_0, gate_1, up_1, ...) → out[M, N/2] = silu(g)*u.
    Optional fused row-scale (norm entry) applied before the activation."""
    if a.is_cuda:
        return _require_ext().gemm256_swiglu(
            a.contiguous(), b_interleaved.contiguous(), stat, stat_mul, stat_eps
        )
    c = a.float() @ b_interleaved.float().t()
    if stat is not None:
        c = c * _rowscale(stat, stat_mul, stat_eps)[:, None]
    g, u = c[:, 0::2], c[:, 1::2]
    return (g * torch.sigmoid(g) * u).to(a.dtype)


def gemm256_resid(
    a: torch.Tensor, b: torch.Tensor, resid: torch.Tensor, want_stat: bool = True
):
    """C = a @ b.T + resid, plus (optionally) the per-row sum of squares of
    C — the next projection's norm statistic — accumulated in the epilogue.
    Returns (C, stat) or (C, None)."""
    if a.is_cuda:
        out = _require_ext().gemm256_resid(
            a.contiguous(), b.contiguous(), resid.contiguous(), want_stat
        )
        return (out[0], out[1]) if want_stat else (out[0], None)
    c = a.float() @ b.float().t() + resid.float()
    stat = c.pow(2).sum(dim=-1) if want_stat else None
    return c.to(a.dtype), stat


def rowsumsq(x: torch.Tensor) -> torch.Tensor:
    """stat[m] = sum_k x[m,k]^2 (f32) over the trailing dim."""
    if x.is_cuda:
        return _require_ext().rowsumsq(x.contiguous())
    k = x.shape[-1]
    return x.float().reshape(-1, k).pow(2).sum(dim=-1)


def gemv_norm(a: torch.Tensor, w: torch.Tensor, mul: float, eps: float) -> torch.Tensor:
    """Decode GEMV with the rmsnorm entry folded in: the kernel computes
    s[m] = rsqrt(sumsq(a_m)*mul + eps) from the x rows it streams anyway
    and scales the dot products — no rmsnorm kernel, no stat buffer."""
    if a.is_cuda:
        return _require_ext().gemv_norm(a.contiguous(), w.contiguous(), mul, eps)
    s = torch.rsqrt(a.float().pow(2).sum(-1) * mul + eps)
    return ((a.float() @ w.float().t()) * s[:, None]).to(a.dtype)


def gemv_resid(a: torch.Tensor, w: torch.Tensor, resid: torch.Tensor) -> torch.Tensor:
    """out = a @ w.T + resid (decode residual add fused into the GEMV)."""
    if a.is_cuda:
        return _require_ext().gemv_resid(a.contiguous(), w.contiguous(), resid.contiguous())
    return (a.float() @ w.float().t() + resid.float()).to(a.dtype)


def gemv_swiglu_norm(a: torch.Tensor, w_interleaved: torch.Tensor, mul: float, eps: float) -> torch.Tensor:
    """Decode gate/up projection with interleaved rows + fused SwiGLU and
    norm entry: out[M, rows/2] = silu(s·g)·(s·u)."""
    if a.is_cuda:
        return _require_ext().gemv_swiglu_norm(a.contiguous(), w_interleaved.contiguous(), mul, eps)
    s = torch.rsqrt(a.float().pow(2).sum(-1) * mul + eps)
    c = (a.float() @ w_interleaved.float().t()) * s[:, None]
    g, u = c[:, 0::2], c[:, 1::2]
    return (g * torch.sigmoid(g) * u).to(a.dtype)


def rope_qkv_decode(qkv2d, kc, vc, inv_freq, pos_dev: torch.Tensor, Hq: int, Hkv: int, D: int):
    """Fused decode head prep: rope q (returned [B,Hq,D]) and rope k +
    copy v straight into the KV cache at DEVICE position pos (hipGraph-
    replayable — one kernel instead of rope + 2 index_copys + 2 copies).
    Rope angles come from inv_freq [D/2] computed in-kernel (no host
    cos/sin table launches)."""
    if qkv2d.is_cuda:
        return _require_ext().rope_qkv_decode(
            qkv2d.contiguous(), kc, vc, inv_freq.contiguous(), pos_dev, Hq, Hkv, D
        )
    # CPU reference: compose from the existing reference ops
    B = qkv2d.shape[0]
    pos = int(pos_dev.item())
    ang = float(pos) * inv_freq.float().reshape(1, -1)
    cos_t, sin_t = torch.cos(ang).expand(B, -1), torch.sin(ang).expand(B, -1)
    q = qkv2d[:, : Hq * D].reshape(B, Hq, D).clone()
    k = qkv2d[:, Hq * D : (Hq + Hkv) * D].reshape(B, Hkv, D).clone()
    v = qkv2d[:, (Hq + Hkv) * D :].reshape(B, Hkv, D)
    q = rope_ref(q, cos_t, sin_t)
    k = rope_ref(k, cos_t, sin_t)
    kc[:, :, pos] = k
    vc[:, :, pos] = v
    return q


def argmax_rows(x: torch.Tensor) -> torch.Tensor:
    """ids[m] = argmax_n x[m,n] (greedy sampling over logits)."""
    if x.is_cuda:
        return _require_ext().argmax_rows(x.contiguous())
    return x.float().argmax(dim=-1)


def attn_decode_t(q, kc, vc, L_dev: torch.Tensor, scale: _t.Optional[float] = None):
    """Graph-capturable decode attention: L read from a device int32 scalar."""
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _require_ext().attn_decode_t(q.contiguous(), kc, vc, L_dev, scale)
    return attn_decode_ref(q, kc, vc, int(L_dev.item()), scale)


def attn_decode(q, kc, vc, L: int, scale: _t.Optional[float] = None):
    scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _require_ext().attn_decode(q.contiguous(), kc, vc, L, scale)
    return attn_decode_ref(q, kc, vc, L, scale)
