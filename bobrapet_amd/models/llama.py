"""Llama-family transformer for the llm-infer engram (MI355X-native).

Random-init weights (the north star allows synthetic payloads / random
weights — BASELINE.json); bf16 end-to-end.  Compute mapping:
  - plain projections → torch.matmul (hipBLASLt on ROCm — library GEMMs)
  - fused hot ops → hand-written CDNA4 kernels (bobrapet_amd.ops):
    fused residual-add+RMSNorm, RoPE, flash attention (MFMA), SiLU-mul
The decode path keeps a [B, Hkv, Smax, D] KV cache per layer resident in
HBM (288 GB/GPU — no paging needed at these sizes).
"""
from __future__ import annotations

import math
import typing as _t
from dataclasses import dataclass

import torch

from .. import ops


@dataclass
class LlamaConfig:
    name: str = "llama-3-8b"
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_seq_len: int = 8192

    @property
    def qkv_out(self) -> int:
        return (self.num_heads + 2 * self.num_kv_heads) * self.head_dim


CONFIGS: _t.Dict[str, LlamaConfig] = {
    "llama-3-8b": LlamaConfig(),
    # small config for tests / CPU smoke
    "llama-tiny": LlamaConfig(
        name="llama-tiny",
        vocab_size=1024,
        hidden_size=256,
        intermediate_size=768,
        num_layers=2,
        num_heads=2,
        num_kv_heads=1,
        head_dim=128,
        max_seq_len=512,
    ),
    # 1B-class config (head_dim held at 128 — the CDNA4 attention kernels
    # are D=128 specialized; 16x128 spans the same hidden width)
    "llama-3-1b": LlamaConfig(
        name="llama-3-1b",
        vocab_size=128256,
        hidden_size=2048,
        intermediate_size=8192,
        num_layers=16,
        num_heads=16,
        num_kv_heads=4,
        head_dim=128,
    ),
    # single-GPU 70B: 141 GB of bf16 weights resident in 288 GB HBM3E —
    # no tensor parallelism needed for b1 serving on one MI355X
    "llama-3-70b": LlamaConfig(
        name="llama-3-70b",
        vocab_size=128256,
        hidden_size=8192,
        intermediate_size=28672,
        num_layers=80,
        num_heads=64,
        num_kv_heads=8,
        head_dim=128,
    ),
}


class LlamaLayerWeights:
    """Layer weights in the MI355X-native serving layout:

    - [out, in] rows (the GEMM/GEMV weight-streaming layout — no
      transposed copies, works at 70B);
    - the RMSNorm per-channel GAINS are FOLDED into w_qkv / w_gate_up
      (rmsnorm(x)·W^T == rowscale(x) · (W·diag(g))^T — the per-row scale
      commutes with the GEMM and is applied in the kernel epilogues, so
      no separate norm kernels run on the hot path);
    - w_gate_up rows are INTERLEAVED (gate_0, up_0, gate_1, up_1, ...) so
      the fused-SwiGLU epilogues pair gate/up in adjacent output columns.
    """
    __slots__ = ("ln_attn", "w_qkv", "w_o", "ln_mlp", "w_gate_up", "w_down")

    def __init__(self, cfg: LlamaConfig, device, dtype, gen):
        H, I = cfg.hidden_size, cfg.intermediate_size
        std = 0.02

        def mk(*shape):
            t = torch.empty(*shape, device=device, dtype=dtype)
            t.normal_(0.0, std, generator=gen)
            return t

        self.ln_attn = torch.ones(H, device=device, dtype=dtype)
        self.ln_mlp = torch.ones(H, device=device, dtype=dtype)
        self.w_qkv = fold_gain(mk(cfg.qkv_out, H), self.ln_attn)
        self.w_o = mk(H, cfg.num_heads * cfg.head_dim)
        gate_up = mk(2 * I, H)
        self.w_gate_up = fold_gain(interleave_gate_up(gate_up), self.ln_mlp)
        self.w_down = mk(H, I)


def fold_gain(w: torch.Tensor, gain: torch.Tensor) -> torch.Tensor:
    """W' = W · diag(g): fold a norm's per-channel gain into the weight."""
    return (w.float() * gain.float()[None, :]).to(w.dtype).contiguous()


def interleave_gate_up(w_gate_up: torch.Tensor) -> torch.Tensor:
    """[gate(I rows); up(I rows)] → rows (g0,u0,g1,u1,...)."""
    I = w_gate_up.shape[0] // 2
    return (
        torch.stack([w_gate_up[:I], w_gate_up[I:]], dim=1)
        .reshape(2 * I, w_gate_up.shape[1])
        .contiguous()
    )


class LlamaModel:
    """Inference-only Llama with explicit weights (no nn.Module overhead)."""

    def __init__(
        self,
        cfg: _t.Union[LlamaConfig, str] = "llama-3-8b",
        device: _t.Union[str, torch.device, None] = None,
        dtype: torch.dtype = torch.bfloat16,
        seed: int = 1234,
    ):
        if isinstance(cfg, str):
            cfg = CONFIGS[cfg]
        self.cfg = cfg
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.dtype = dtype
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed)
        std = 0.02
        self.embed = torch.empty(
            cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=dtype
        ).normal_(0.0, std, generator=gen)
        self.layers = [
            LlamaLayerWeights(cfg, self.device, dtype, gen) for _ in range(cfg.num_layers)
        ]
        self.ln_final = torch.ones(cfg.hidden_size, device=self.device, dtype=dtype)
        self.lm_head = fold_gain(
            torch.empty(
                cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=dtype
            ).normal_(0.0, std, generator=gen),
            self.ln_final,
        )
        self._inv_h = 1.0 / cfg.hidden_size
        if self.device.type == "cuda":
            # the fused-epilogue GEMM path needs these alignments; every
            # shipped config satisfies them (gemm256: N%256, K%32)
            for n, k in (
                (cfg.qkv_out, cfg.hidden_size),
                (cfg.hidden_size, cfg.num_heads * cfg.head_dim),
                (2 * cfg.intermediate_size, cfg.hidden_size),
                (cfg.hidden_size, cfg.intermediate_size),
                (cfg.vocab_size, cfg.hidden_size),
            ):
                assert n % 256 == 0 and k % 32 == 0, (
                    f"config {cfg.name}: projection [{n},{k}] violates the "
                    "gemm256 alignment (N%256, K%32)"
                )
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        self._kv_cache: _t.Optional[_t.List[_t.Tuple[torch.Tensor, torch.Tensor]]] = None
        self._cache_len = 0
        self._decode_graph = None
        self._g_ids = None  # static graph buffers
        self._g_pos = None
        self._g_logits = None
        per_layer = (
            2 * cfg.hidden_size
            + cfg.hidden_size * cfg.qkv_out
            + cfg.num_heads * cfg.head_dim * cfg.hidden_size
            + 3 * cfg.hidden_size * cfg.intermediate_size
        )
        n_params = (
            2 * cfg.vocab_size * cfg.hidden_size + cfg.num_layers * per_layer + cfg.hidden_size
        )
        self.param_bytes = n_params * self.embed.element_size()

    # ------------------------------------------------------------------

    def param_count(self) -> int:
        cfg = self.cfg
        per_layer = (
            cfg.hidden_size * cfg.qkv_out
            + cfg.num_heads * cfg.head_dim * cfg.hidden_size
            + cfg.hidden_size * 2 * cfg.intermediate_size
            + cfg.intermediate_size * cfg.hidden_size
            + 2 * cfg.hidden_size
        )
        return (
            cfg.vocab_size * cfg.hidden_size * 2
            + cfg.num_layers * per_layer
            + cfg.hidden_size
        )

    def _split_qkv(self, qkv: torch.Tensor, B: int, S: int):
        cfg = self.cfg
        nq = cfg.num_heads * cfg.head_dim
        nkv = cfg.num_kv_heads * cfg.head_dim
        q = qkv[..., :nq].view(B, S, cfg.num_heads, cfg.head_dim)
        k = qkv[..., nq : nq + nkv].view(B, S, cfg.num_kv_heads, cfg.head_dim)
        v = qkv[..., nq + nkv :].view(B, S, cfg.num_kv_heads, cfg.head_dim)
        return q, k, v

    # ------------------------------------------------------------------

    @torch.no_grad()
    def prefill(
        self,
        ids: torch.Tensor,
        logits_for_all: bool = False,
        fill_cache: bool = False,
        seq_shard: _t.Optional[_t.Tuple[int, int]] = None,
    ) -> torch.Tensor:
        """ids [B, S] int64 → logits [B, V] (last position) or [B, S, V].

        seq_shard=(rank, world) runs SEQUENCE-PARALLEL prefill: `ids` is
        this rank's contiguous shard of a rank-major-partitioned global
        sequence (weights replicated), attention is ring attention over
        xGMI (parallel/ring_attention.py), rope positions are offset by the
        shard start.  Returns this rank's local logits."""
        cfg = self.cfg
        B, S = ids.shape
        pos0 = 0
        ring = None
        if seq_shard is not None and seq_shard[1] > 1:
            from ..parallel.ring_attention import ring_attention as ring

            pos0 = seq_shard[0] * S
        positions = torch.arange(pos0, pos0 + S, device=self.device)
        cos_t, sin_t = ops.rope_tables(positions, cfg.head_dim, cfg.rope_theta)
        # tables are per flattened token (T = B*S)
        cos_f = cos_t.repeat(B, 1)
        sin_f = sin_t.repeat(B, 1)

        # the residual stream x2 [M=B*S, H] and its per-row sumsq `stat`
        # flow through FUSED-epilogue GEMMs (csrc/hip/gemm256.hip): the
        # rmsnorm entries become row scales inside the projections (gains
        # folded into the weights), SwiGLU runs in the gate/up epilogue,
        # and the residual adds + next-norm statistics come out of the
        # o/down projection epilogues — no separate norm/activation/add
        # kernels and no normalized-x round trips through HBM.
        M = B * S
        x2 = self.embed[ids.to(self.device)].reshape(M, cfg.hidden_size)
        stat = ops.rowsumsq(x2)
        if fill_cache:
            self._alloc_cache(B, max(cfg.max_seq_len, S))

        fuse_qrope = ring is None and x2.is_cuda
        if fuse_qrope:
            self._ensure_inv_freq()
        nq = cfg.num_heads * cfg.head_dim
        nkv = cfg.num_kv_heads * cfg.head_dim
        for li, lw in enumerate(self.layers):
            qkv = ops.gemm256_nt(x2, lw.w_qkv, stat, self._inv_h, cfg.rms_eps)
            if fuse_qrope:
                # rope only K (Q is roped on load inside the attention
                # kernel); q and v are STRIDED views into qkv — neither
                # round-trips HBM through a rope/copy kernel
                kh = ops.rope_k_from_qkv(
                    qkv, cfg.num_heads, cfg.num_kv_heads, cfg.head_dim,
                    cos_f, sin_f,
                ).view(B, S, cfg.num_kv_heads, cfg.head_dim)
                q3 = qkv.view(B, S, -1)
                qh = q3[..., :nq].unflatten(-1, (cfg.num_heads, cfg.head_dim))
                vh = q3[..., nq + nkv:].unflatten(
                    -1, (cfg.num_kv_heads, cfg.head_dim))
            else:
                qh, kh, vh = ops.rope_qkv_split(
                    qkv.view(B, S, -1), B, S, cfg.num_heads, cfg.num_kv_heads,
                    cfg.head_dim, cos_f, sin_f,
                )
            if fill_cache:
                kc, vc = self._kv_cache[li]
                kc[:, :, :S] = kh.transpose(1, 2)
                vc[:, :, :S] = vh.transpose(1, 2)
            if ring is not None:
                attn = ring(qh, kh, vh, self.scale, causal=True)
            elif fuse_qrope:
                attn = ops.attn_prefill_qrope(
                    qh, kh, vh, self._inv_freq, pos0, self.scale, causal=True)
            else:
                attn = ops.attn_prefill(qh, kh, vh, self.scale, causal=True)
            a2 = attn.reshape(M, cfg.num_heads * cfg.head_dim)
            x2, stat = ops.gemm256_resid(a2, lw.w_o, x2)
            act = ops.gemm256_swiglu(x2, lw.w_gate_up, stat, self._inv_h, cfg.rms_eps)
            x2, stat = ops.gemm256_resid(act, lw.w_down, x2)

        if fill_cache:
            self._cache_len = S
        if logits_for_all:
            logits = ops.gemm256_nt(x2, self.lm_head, stat, self._inv_h, cfg.rms_eps)
            return logits.view(B, S, cfg.vocab_size)
        xl = x2.view(B, S, cfg.hidden_size)[:, -1].contiguous()
        if B <= 8 and self.device.type == "cuda":
            # tiny-M logits: the weight-streaming GEMV (norm entry fused)
            # beats a 256-row-tile GEMM at M=B
            return ops.gemv_norm(xl, self.lm_head, self._inv_h, cfg.rms_eps)
        sl = stat.view(B, S)[:, -1].contiguous()
        return ops.gemm256_nt(xl, self.lm_head, sl, self._inv_h, cfg.rms_eps)

    # ------------------------------------------------------------------

    def _ensure_inv_freq(self) -> None:
        if not hasattr(self, "_inv_freq"):
            cfg = self.cfg
            self._inv_freq = 1.0 / (
                cfg.rope_theta
                ** (
                    torch.arange(0, cfg.head_dim, 2, dtype=torch.float32, device=self.device)
                    / cfg.head_dim
                )
            )

    def _decode_body(self, ids: torch.Tensor, pos_i32: torch.Tensor) -> torch.Tensor:
        """One decode step with all dynamic state in device tensors — every
        op here is hipGraph-capturable (no host-dependent shapes/values).

        B<=8 runs entirely on the fused weight-streaming GEMVs
        (csrc/hip/gemm.hip gemv2): the rmsnorm entries are computed from
        the x rows the GEMV streams anyway (norm gains folded into the
        weights), residual adds and SwiGLU live in the GEMV epilogues —
        6 kernels per layer, no elementwise passes."""
        cfg = self.cfg
        B = ids.shape[0]
        inv_freq = self._inv_freq
        on_cuda = self.device.type == "cuda"
        # decode compute tiers (all hand-written kernels, no at::native):
        #   B <= 8   weight-streaming fused GEMVs
        #   B <= 32  skinny-M MFMA GEMM (gemmsk: N-tile x K-split grid)
        #   B >  32  the prefill gemm256 chain (256-row tiles amortize)
        use_gemv = B <= 8 and on_cuda
        use_sk = (not use_gemv) and B <= 32 and on_cuda
        use_gemm = (not use_gemv) and (not use_sk) and on_cuda
        cos_t = sin_t = None
        pos_l = None
        if not on_cuda:
            # CPU reference path builds host-side tables
            pos_l = pos_i32.to(torch.long)
            ang = pos_l.float().reshape(1, 1) * inv_freq[None, :]
            cos_t = torch.cos(ang).expand(B, -1).contiguous()
            sin_t = torch.sin(ang).expand(B, -1).contiguous()

        x2 = self.embed[ids].view(B, cfg.hidden_size)  # residual stream
        stat = ops.rowsumsq(x2) if (use_gemm or use_sk) else None
        L_dev = (pos_i32 + 1).contiguous()
        for li, lw in enumerate(self.layers):
            kc, vc = self._kv_cache[li]
            if use_gemv:
                qkv = ops.gemv_norm(x2, lw.w_qkv, self._inv_h, cfg.rms_eps)
            elif use_sk:
                qkv = ops.gemmsk_nt(x2, lw.w_qkv, stat, self._inv_h, cfg.rms_eps)
            elif use_gemm:
                qkv = ops.gemm256_nt(x2, lw.w_qkv, stat, self._inv_h, cfg.rms_eps)
            else:
                qkv = ops.gemv_norm(x2, lw.w_qkv, self._inv_h, cfg.rms_eps)
            if on_cuda:
                # fused rope + cache append: one kernel instead of rope +
                # two index_copys + two layout copies (x32 layers/step)
                qf = ops.rope_qkv_decode(
                    qkv, kc, vc, inv_freq, pos_i32,
                    cfg.num_heads, cfg.num_kv_heads, cfg.head_dim,
                )
            else:
                q, k, v = self._split_qkv(qkv.view(B, 1, -1), B, 1)
                qf = q.reshape(B, cfg.num_heads, cfg.head_dim).contiguous()
                kf = k.reshape(B, cfg.num_kv_heads, cfg.head_dim).contiguous()
                qf, kf = ops.rope_inplace(qf, kf, cos_t, sin_t)
                # device-indexed cache append (graph-replayable)
                kc.index_copy_(2, pos_l, kf.view(B, cfg.num_kv_heads, 1, cfg.head_dim))
                vc.index_copy_(2, pos_l, v.reshape(B, cfg.num_kv_heads, 1, cfg.head_dim).contiguous())
            attn = ops.attn_decode_t(qf, kc, vc, L_dev, self.scale)
            a2 = attn.reshape(B, cfg.num_heads * cfg.head_dim)
            if use_sk:
                x2, stat = ops.gemmsk_resid(a2, lw.w_o, x2)
                act = ops.gemmsk_swiglu(x2, lw.w_gate_up, stat, self._inv_h, cfg.rms_eps)
                x2, stat = ops.gemmsk_resid(act, lw.w_down, x2)
            elif use_gemm:
                x2, stat = ops.gemm256_resid(a2, lw.w_o, x2)
                act = ops.gemm256_swiglu(x2, lw.w_gate_up, stat, self._inv_h, cfg.rms_eps)
                x2, stat = ops.gemm256_resid(act, lw.w_down, x2)
            else:
                x2 = ops.gemv_resid(a2, lw.w_o, x2)
                act = ops.gemv_swiglu_norm(x2, lw.w_gate_up, self._inv_h, cfg.rms_eps)
                x2 = ops.gemv_resid(act, lw.w_down, x2)
        if use_sk:
            return ops.gemmsk_nt(x2, self.lm_head, stat, self._inv_h, cfg.rms_eps)
        if use_gemm:
            return ops.gemm256_nt(x2, self.lm_head, stat, self._inv_h, cfg.rms_eps)
        return ops.gemv_norm(x2, self.lm_head, self._inv_h, cfg.rms_eps)

    def decode_step_graphed(self, ids: torch.Tensor) -> torch.Tensor:
        """hipGraph-captured decode (guide: capture launch-bound inner loops
        in hipGraphs): one replay per token; position/length live in device
        scalars that the graph itself advances.  Graphs are cached per
        batch size (a served model alternates batches), and the device
        position is re-synced whenever a new prefill reset the cache — so a
        process-wide cached model serves many sequential stories safely."""
        assert self._kv_cache is not None, "call prefill(fill_cache=True) first"
        B = ids.shape[0]
        self._ensure_inv_freq()
        if not hasattr(self, "_graphs"):
            self._graphs = {}
        entry = self._graphs.get(B)
        if entry is None or entry["cache_id"] != id(self._kv_cache):
            # capture for this batch (or re-capture: prefill reallocated the
            # KV cache, so the captured pointers are stale)
            g_ids = torch.zeros(B, dtype=torch.long, device=self.device)
            g_pos = torch.zeros(1, dtype=torch.int32, device=self.device)
            g_ids.copy_(ids)
            g_pos.fill_(self._cache_len)
            # warmup outside capture (allocator settles)
            self._decode_body(g_ids, g_pos)
            g_pos.fill_(self._cache_len)
            torch.cuda.synchronize(self.device)
            g = torch.cuda.CUDAGraph()
            stream = torch.cuda.Stream(device=self.device)
            with torch.cuda.stream(stream):
                with torch.cuda.graph(g, stream=stream):
                    g_logits = self._decode_body(g_ids, g_pos)
                    g_pos.add_(1)  # the graph advances its own position
            entry = {
                "graph": g, "ids": g_ids, "pos": g_pos, "logits": g_logits,
                "pos_val": self._cache_len, "cache_id": id(self._kv_cache),
            }
            self._graphs[B] = entry
        if entry["pos_val"] != self._cache_len:
            # a new prefill moved the cache length since the last replay
            entry["pos"].fill_(self._cache_len)
            entry["pos_val"] = self._cache_len
        entry["ids"].copy_(ids)
        entry["graph"].replay()
        entry["pos_val"] += 1
        self._cache_len += 1
        return entry["logits"]

    def _alloc_cache(self, B: int, smax: int) -> None:
        cfg = self.cfg
        if (
            self._kv_cache is not None
            and self._kv_cache[0][0].shape[0] == B
            and self._kv_cache[0][0].shape[2] >= smax
        ):
            # reuse the allocation: decode never reads beyond _cache_len, so
            # no zeroing needed — and captured decode graphs keep pointing
            # at live tensors across stories
            self._cache_len = 0
            return
        self._kv_cache = [
            (
                torch.zeros(
                    B, cfg.num_kv_heads, smax, cfg.head_dim, device=self.device, dtype=self.dtype
                ),
                torch.zeros(
                    B, cfg.num_kv_heads, smax, cfg.head_dim, device=self.device, dtype=self.dtype
                ),
            )
            for _ in range(cfg.num_layers)
        ]
        self._cache_len = 0

    @torch.no_grad()
    def decode_step(self, ids: torch.Tensor) -> torch.Tensor:
        """One token per sequence: ids [B] → logits [B, V]; uses the cache
        filled by prefill(fill_cache=True).  Eager (ungraphed) path —
        same body as the graphed one."""
        assert self._kv_cache is not None, "call prefill(fill_cache=True) first"
        self._ensure_inv_freq()
        pos_i32 = torch.tensor([self._cache_len], dtype=torch.int32, device=self.device)
        logits = self._decode_body(ids.to(self.device), pos_i32)
        self._cache_len += 1
        return logits

    @torch.no_grad()
    def generate(
        self, ids: torch.Tensor, new_tokens: int, greedy: bool = True,
        use_graph: _t.Optional[bool] = None,
    ):
        """Prefill + decode loop; returns generated token ids [B, new_tokens].
        On GPU the decode loop replays a captured hipGraph per token."""
        if use_graph is None:
            use_graph = self.device.type == "cuda"
        logits = self.prefill(ids, fill_cache=True)
        step = self.decode_step_graphed if use_graph else self.decode_step
        out = []
        for _ in range(new_tokens):
            nxt = ops.argmax_rows(logits) if greedy else torch.multinomial(
                torch.softmax(logits.float(), dim=-1), 1
            ).squeeze(-1)
            out.append(nxt)
            logits = step(nxt)
        return torch.stack(out, dim=1)


_MODEL_CACHE: _t.Dict[_t.Tuple[str, str], LlamaModel] = {}


def get_model(name: str = "llama-3-8b", device=None) -> LlamaModel:
    """Process-wide model cache (weights stay resident in HBM between steps)."""
    dev = str(device or ("cuda" if torch.cuda.is_available() else "cpu"))
    key = (name, dev)
    if key not in _MODEL_CACHE:
        _MODEL_CACHE[key] = LlamaModel(name, device=dev)
    return _MODEL_CACHE[key]
