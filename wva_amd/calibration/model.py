"""MI355X-native Llama-family decode engine for capacity calibration.

The reference's capacity model runs on parameters fit OFFLINE from vLLM
benchmarks on NVIDIA/MI300X hardware (docs/design/modeling-optimization.md
:46-84). This module is the MI355X-native replacement: a real bf16 decode
step of the target architecture (random-init weights — no network for
checkpoints) built from this repo's HIP/CDNA4 kernels (wva_amd.ops) with
GEMMs on hipBLASLt via torch.matmul, used to MEASURE ITL-vs-batch curves
and KV capacity on the actual GPU.

Single-GPU by design: TP over RCCL/xGMI shards these same GEMMs; the
autoscaler consumes per-(model, gpuCount) profiles, measured per TP degree
(capacity_store keys records by gpu_count).
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional

import torch

from .. import ops


@dataclass
class LlamaConfig:
    name: str = "llama"
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_q_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    vocab_size: int = 128256
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5

    @property
    def q_size(self) -> int:
        return self.num_q_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim

    def kv_bytes_per_token(self, dtype_bytes: int = 2) -> int:
        return 2 * self.num_layers * self.kv_size * dtype_bytes

    def weight_bytes(self, dtype_bytes: int = 2) -> int:
        per_layer = (
            self.hidden_size * (self.q_size + 2 * self.kv_size)  # qkv
            + self.q_size * self.hidden_size  # o
            + 3 * self.hidden_size * self.intermediate_size  # gate/up/down
            + 2 * self.hidden_size  # norms
        )
        return dtype_bytes * (
            self.num_layers * per_layer
            + 2 * self.vocab_size * self.hidden_size  # embed + lm_head
            + self.hidden_size
        )


LLAMA_3_8B = LlamaConfig(
    name="meta-llama/Llama-3.1-8B",
    hidden_size=4096,
    intermediate_size=14336,
    num_layers=32,
    num_q_heads=32,
    num_kv_heads=8,
)

LLAMA_3_70B = LlamaConfig(
    name="meta-llama/Llama-3.1-70B",
    hidden_size=8192,
    intermediate_size=28672,
    num_layers=80,
    num_q_heads=64,
    num_kv_heads=8,
)

# Tiny config for smoke tests (same topology, minutes → milliseconds)
TINY = LlamaConfig(
    name="tiny-llama",
    hidden_size=1024,
    intermediate_size=2816,
    num_layers=2,
    num_q_heads=8,
    num_kv_heads=2,
    vocab_size=32000,
)


def _quantize_fp8(w: torch.Tensor):
    """Per-channel (rowwise) e4m3 quantization: (w_fp8 [N,K], scale [N]).
    Row scales commute with the GEMM's k-sum, so they fold into the
    output store exactly (csrc/skinny_gemm.hip W_FP8)."""
    scale = (w.abs().amax(dim=1).float() / 448.0).clamp_min(1e-12)  # [N]
    w8 = (
        (w.float() / scale[:, None]).clamp(-448.0, 448.0)
        .to(torch.float8_e4m3fn)
    )
    return w8.contiguous(), scale.contiguous()


class _DecoderLayer:
    def __init__(self, cfg: LlamaConfig, device, dtype, gen):
        h, std = cfg.hidden_size, 0.02
        def w(rows, cols):
            return torch.empty(rows, cols, device=device, dtype=dtype).normal_(
                0.0, std, generator=gen
            )

        self.input_norm = torch.ones(h, device=device, dtype=dtype)
        self.post_attn_norm = torch.ones(h, device=device, dtype=dtype)
        self.wqkv = w(cfg.q_size + 2 * cfg.kv_size, h)
        self.wo = w(h, cfg.q_size)
        self.w_gate_up = w(2 * cfg.intermediate_size, h)
        self.w_down = w(h, cfg.intermediate_size)


class LlamaDecodeModel:
    """Decode-only engine with a contiguous KV cache [B, S, Hk, D]."""

    def __init__(
        self,
        cfg: LlamaConfig,
        max_batch: int = 256,
        max_seq: int = 2048,
        device: str = "cuda",
        seed: int = 0,
        kv_dtype: str = "bf16",
        weights_dtype: str = "bf16",
    ):
        self.cfg = cfg
        self.device = torch.device(device)
        self.dtype = torch.bfloat16
        # fp8 weights: per-tensor e4m3 quantization, GEMMs via
        # torch._scaled_mm (W8A8 with dynamic per-tensor activation
        # scales) — the quantized-serving configuration (vLLM --dtype
        # fp8 analog); halves GEMM weight streaming. Calibration-variant
        # only: the headline bench stays bf16 weights.
        if weights_dtype not in ("bf16", "fp8"):
            raise ValueError(f"weights_dtype must be bf16|fp8")
        self.weights_dtype = weights_dtype
        # fp8 (e4m3) KV storage: half the attention HBM traffic, double
        # the KV capacity; compute stays bf16 (kernels up-convert at
        # fragment build — see csrc/attention.hip KV_FP8)
        if kv_dtype not in ("bf16", "fp8"):
            raise ValueError(f"kv_dtype must be bf16|fp8, got {kv_dtype}")
        self.kv_dtype = kv_dtype
        self.cache_dtype = (
            torch.float8_e4m3fn if kv_dtype == "fp8" else torch.bfloat16
        )
        self.max_batch = max_batch
        self.max_seq = max_seq
        gen = torch.Generator(device=self.device)
        gen.manual_seed(seed)

        self.embed = torch.empty(
            cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=self.dtype
        ).normal_(0.0, 0.02, generator=gen)
        self.layers: List[_DecoderLayer] = [
            _DecoderLayer(cfg, self.device, self.dtype, gen)
            for _ in range(cfg.num_layers)
        ]
        self.final_norm = torch.ones(
            cfg.hidden_size, device=self.device, dtype=self.dtype
        )
        self.lm_head = torch.empty(
            cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=self.dtype
        ).normal_(0.0, 0.02, generator=gen)
        if weights_dtype == "fp8":
            for layer in self.layers:
                for attr in ("wqkv", "wo", "w_gate_up", "w_down"):
                    setattr(layer, attr + "_q",
                            _quantize_fp8(getattr(layer, attr)))
                    setattr(layer, attr, None)  # free the bf16 copy
            self.lm_head_q = _quantize_fp8(self.lm_head)
            self.lm_head = None

        # Head-major contiguous KV cache per layer: [B, Hk, S, D] — each
        # (sequence, kv-head) is one sequential HBM stream for the
        # attention kernel's tile staging (docs/mi355x-kernels.md)
        self.k_cache = [
            torch.zeros(
                max_batch, cfg.num_kv_heads, max_seq, cfg.head_dim,
                device=self.device, dtype=self.cache_dtype,
            )
            for _ in range(cfg.num_layers)
        ]
        self.v_cache = [torch.zeros_like(self.k_cache[0]) for _ in range(cfg.num_layers)]
        self.context_lens = torch.zeros(
            max_batch, dtype=torch.int32, device=self.device
        )
        self.scale = 1.0 / math.sqrt(cfg.head_dim)

    def reset(self, batch: int, context_len: int) -> None:
        """Random-fill KV up to context_len for `batch` sequences."""
        self.context_lens.zero_()
        self.context_lens[:batch] = context_len
        for layer in range(self.cfg.num_layers):
            if self.cache_dtype == torch.bfloat16:
                self.k_cache[layer][:batch, :, :context_len].normal_(0.0, 1.0)
                self.v_cache[layer][:batch, :, :context_len].normal_(0.0, 1.0)
            else:  # normal_ unsupported on float8: generate + cast
                shape = self.k_cache[layer][:batch, :, :context_len].shape
                for cache in (self.k_cache, self.v_cache):
                    cache[layer][:batch, :, :context_len] = torch.randn(
                        shape, device=self.device, dtype=torch.bfloat16
                    ).to(self.cache_dtype)

    def _linear(self, x: torch.Tensor, layer, name: str) -> torch.Tensor:
        """bf16: ops.linear (hipBLASLt); fp8 weights: weight-only W8A16
        on the in-tree streaming kernel (activations stay bf16; weights
        up-convert in-fragment; per-channel scales fold into the store).
        W8A8 via torch._scaled_mm measured 2.4× slower at decode sizes
        (dynamic-quant launch overhead — docs/mi355x-kernels.md), so the
        weight-only form is the fp8 GEMM path."""
        if self.weights_dtype == "bf16":
            w = getattr(layer, name) if layer is not None else self.lm_head
            return ops.linear(x, w)
        w8, w_scale = (
            getattr(layer, name + "_q") if layer is not None else self.lm_head_q
        )
        if not x.is_cuda:  # CPU reference: dequantized matmul
            return (
                x.float() @ (w8.float() * w_scale[:, None]).t()
            ).to(x.dtype)
        if x.shape[0] <= 64 and x.shape[1] % 128 == 0:
            from .. import ops as _ops

            return _ops._require_ext().skinny_linear_fp8(x, w8, w_scale)
        # large-M fallback (prefill-scale): dequantize and use hipBLASLt
        return (x @ (w8.to(x.dtype) * w_scale[:, None].to(x.dtype)).t())

    @torch.no_grad()
    def decode_step(self, token_ids: torch.Tensor) -> torch.Tensor:
        """One decode iteration for `B = len(token_ids)` sequences.
        Appends each sequence's new KV at position context_lens[b] and
        returns logits [B, vocab]."""
        cfg = self.cfg
        B = token_ids.shape[0]
        positions = self.context_lens[:B].clone()

        ctx = positions + 1  # includes the new token (hoisted: one
        # launch per step, not per layer — profiling showed 32 redundant
        # int-add launches/step, profiles/decode8b_r2_kernel_stats.txt)
        x = self.embed.index_select(0, token_ids)  # [B, H]
        residual: Optional[torch.Tensor] = None

        for li, layer in enumerate(self.layers):
            if residual is None:
                residual = x.clone()
                h = ops.rmsnorm(x, layer.input_norm, None, cfg.rms_eps)
            else:
                h = ops.rmsnorm(x, layer.input_norm, residual, cfg.rms_eps)

            qkv = self._linear(h, layer, "wqkv")
            # fused: RoPE on the strided qkv row + KV-cache append
            q = ops.rope_append_kv(
                qkv, self.k_cache[li][:B], self.v_cache[li][:B],
                positions, cfg.num_q_heads, cfg.num_kv_heads, cfg.rope_theta,
            )

            attn = ops.gqa_decode_attn(
                q, self.k_cache[li][:B], self.v_cache[li][:B], ctx, self.scale
            )
            x = self._linear(attn.reshape(B, cfg.q_size), layer, "wo")

            h2 = ops.rmsnorm(x, layer.post_attn_norm, residual, cfg.rms_eps)
            gate_up = self._linear(h2, layer, "w_gate_up")
            act = ops.silu_mul_fused(gate_up)
            x = self._linear(act, layer, "w_down")

        final = ops.rmsnorm(x, self.final_norm, residual, cfg.rms_eps)
        logits = self._linear(final, None, "lm_head")
        self.context_lens[:B] += 1
        return logits

    @torch.no_grad()
    def prefill(self, token_ids: torch.Tensor) -> torch.Tensor:
        """Prefill `token_ids [B, S]`: computes all S positions in one
        pass (causal attention), fills the KV cache, sets context_lens=S
        and returns last-position logits [B, vocab].

        Grounds the ServiceProfile's prefill_tokens_per_s (the TTFT
        model's input) with a measured value instead of a constant.
        Attention here is explicit matmul attention on hipBLASLt —
        prefill is GEMM-shaped compute (S×S scores), exactly what the
        library is for; the hand-written kernels cover the memory-bound
        decode path. RMSNorm/RoPE/SwiGLU reuse the HIP kernels (they are
        row-shaped and batch-size-agnostic).
        """
        cfg = self.cfg
        if self.weights_dtype != "bf16":
            raise NotImplementedError(
                "prefill with fp8 weights is not implemented (the "
                "weight-only fp8 mode targets decode; prefill would "
                "dequantize every layer per pass)"
            )
        B, S = token_ids.shape
        T = B * S
        if S > self.max_seq:
            raise ValueError(f"S={S} exceeds max_seq={self.max_seq}")
        positions = (
            torch.arange(S, device=self.device, dtype=torch.int32)
            .repeat(B)
        )  # [T] per-token position

        x = self.embed.index_select(0, token_ids.reshape(-1))  # [T, H]
        residual: Optional[torch.Tensor] = None

        causal = torch.full(
            (S, S), float("-inf"), device=self.device, dtype=torch.float32
        ).triu(1)

        for li, layer in enumerate(self.layers):
            if residual is None:
                residual = x.clone()
                h = ops.rmsnorm(x, layer.input_norm, None, cfg.rms_eps)
            else:
                h = ops.rmsnorm(x, layer.input_norm, residual, cfg.rms_eps)

            qkv = h @ layer.wqkv.t()  # [T, (Hq+2Hk)·D] — prefill M is large
            q, k, v = qkv.split(
                [cfg.q_size, cfg.kv_size, cfg.kv_size], dim=-1
            )
            q = q.reshape(T, cfg.num_q_heads, cfg.head_dim).contiguous()
            k = k.reshape(T, cfg.num_kv_heads, cfg.head_dim).contiguous()
            ops.rope(q, k, positions, cfg.rope_theta)  # in-place HIP kernel
            v = v.reshape(T, cfg.num_kv_heads, cfg.head_dim)

            # append to the head-major cache [B, Hk, S_max, D]
            k_b = k.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
            v_b = v.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
            self.k_cache[li][:B, :, :S].copy_(k_b.permute(0, 2, 1, 3))
            self.v_cache[li][:B, :, :S].copy_(v_b.permute(0, 2, 1, 3))

            if q.is_cuda:
                # hand-written MFMA flash-prefill kernel (gfx950)
                attn = ops.prefill_attn(
                    q, self.k_cache[li], self.v_cache[li], B, S, self.scale
                ).reshape(T, cfg.q_size)
            else:
                # CPU numerics reference: explicit causal matmul attention
                G = cfg.num_q_heads // cfg.num_kv_heads
                qh = (
                    q.reshape(B, S, cfg.num_kv_heads, G, cfg.head_dim)
                    .permute(0, 2, 3, 1, 4)
                    .reshape(B * cfg.num_kv_heads * G, S, cfg.head_dim)
                )
                kh = (
                    self.k_cache[li][:B, :, :S].to(self.dtype)
                    .unsqueeze(2)
                    .expand(B, cfg.num_kv_heads, G, S, cfg.head_dim)
                    .reshape(B * cfg.num_kv_heads * G, S, cfg.head_dim)
                )
                vh = (
                    self.v_cache[li][:B, :, :S].to(self.dtype)
                    .unsqueeze(2)
                    .expand(B, cfg.num_kv_heads, G, S, cfg.head_dim)
                    .reshape(B * cfg.num_kv_heads * G, S, cfg.head_dim)
                )
                scores = (
                    torch.bmm(qh, kh.transpose(1, 2)).float() * self.scale
                    + causal
                )
                p = torch.softmax(scores, dim=-1).to(self.dtype)
                attn = torch.bmm(p, vh)  # [B*Hq, S, D]
                attn = (
                    attn.reshape(B, cfg.num_q_heads, S, cfg.head_dim)
                    .permute(0, 2, 1, 3)
                    .reshape(T, cfg.q_size)
                ).contiguous()
            x = attn @ layer.wo.t()

            h2 = ops.rmsnorm(x, layer.post_attn_norm, residual, cfg.rms_eps)
            gate_up = h2 @ layer.w_gate_up.t()
            act = ops.silu_mul_fused(gate_up)
            x = act @ layer.w_down.t()

        final = ops.rmsnorm(x, self.final_norm, residual, cfg.rms_eps)
        last = final.reshape(B, S, cfg.hidden_size)[:, -1].contiguous()
        logits = last @ self.lm_head.t()
        self.context_lens.zero_()
        self.context_lens[:B] = S
        return logits
