"""Mixtral-family (sparse MoE) decode engine for MI355X calibration.

BASELINE config #4 pairs Llama-3-8B with Mixtral-8×7B under service-class
priorities; this engine provides the measured service profile for the MoE
family the same way calibration/model.py does for dense Llama. Mixtral
8×7B bf16 is ≈ 94 GB of weights — resident on a single 288 GB MI355X (a
192 GB MI300X needs TP=2), which is itself an MI355X capacity statement
the cost solver exploits.

Architecture: Llama-style attention (same HIP kernels: fused add-RMSNorm,
rope_append_kv, GQA decode attention) + top-2 routed expert MLPs. At
decode batches the expert MLPs run DENSE-batched — one hipBLASLt
strided-batched GEMM pair per layer over all experts (see _moe_mlp) —
because decode MoE is weight-streaming-bound and the sparse gather path's
per-expert `nonzero()` syncs dominate.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import List, Optional

import torch

from .. import ops


@dataclass
class MixtralConfig:
    name: str = "mistralai/Mixtral-8x7B"
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_q_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    vocab_size: int = 32000
    rope_theta: float = 1000000.0
    rms_eps: float = 1e-5
    num_experts: int = 8
    top_k: int = 2

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
            self.hidden_size * (self.q_size + 2 * self.kv_size)
            + self.q_size * self.hidden_size
            + self.num_experts * 3 * self.hidden_size * self.intermediate_size
            + self.hidden_size * self.num_experts  # router
            + 2 * self.hidden_size
        )
        return dtype_bytes * (
            self.num_layers * per_layer
            + 2 * self.vocab_size * self.hidden_size
            + self.hidden_size
        )


MIXTRAL_8X7B = MixtralConfig()

TINY_MOE = MixtralConfig(
    name="tiny-mixtral",
    hidden_size=512,
    intermediate_size=1024,
    num_layers=2,
    num_q_heads=4,
    num_kv_heads=1,
    vocab_size=1000,
    num_experts=4,
    top_k=2,
)


class _MoELayer:
    def __init__(self, cfg: MixtralConfig, device, dtype, gen):
        h, std = cfg.hidden_size, 0.02

        def w(rows, cols):
            return torch.empty(rows, cols, device=device, dtype=dtype).normal_(
                0.0, std, generator=gen
            )

        self.input_norm = torch.ones(h, device=device, dtype=dtype)
        self.post_attn_norm = torch.ones(h, device=device, dtype=dtype)
        self.wqkv = w(cfg.q_size + 2 * cfg.kv_size, h)
        self.wo = w(h, cfg.q_size)
        self.w_router = w(cfg.num_experts, h)
        # experts as STACKED tensors [E, 2I, H] / [E, H, I]: the decode
        # path runs them as one strided-batched GEMM pair per layer (see
        # MixtralDecodeModel._moe_mlp); per-expert views kept for the
        # sparse reference path
        self.w_gate_up_stacked = torch.stack(
            [w(2 * cfg.intermediate_size, h) for _ in range(cfg.num_experts)]
        )
        self.w_down_stacked = torch.stack(
            [w(h, cfg.intermediate_size) for _ in range(cfg.num_experts)]
        )
        self.w_gate_up = list(self.w_gate_up_stacked)
        self.w_down = list(self.w_down_stacked)
        # dense-path operands (no transposed VIEWS at run time — hipBLASLt
        # takes a catastrophically slow route for strided-batched GEMMs on
        # transposed views of these shapes; measured 11 ms/layer):
        #  * gate_up runs as ONE flat GEMM against [E·2I, H] (a free view)
        #  * down runs as bmm against a pre-transposed CONTIGUOUS [E, I, H]
        self.w_gate_up_flat = self.w_gate_up_stacked.reshape(
            cfg.num_experts * 2 * cfg.intermediate_size, h
        )
        self.w_down_t = self.w_down_stacked.transpose(1, 2).contiguous()

    def set_expert_weights(self, gate_up_stacked, down_stacked) -> None:
        """Replace expert weights, recomputing every derived operand
        (tests copy weights between devices through this)."""
        self.w_gate_up_stacked = gate_up_stacked
        self.w_down_stacked = down_stacked
        self.w_gate_up = list(gate_up_stacked)
        self.w_down = list(down_stacked)
        e2i, h = (
            gate_up_stacked.shape[0] * gate_up_stacked.shape[1],
            gate_up_stacked.shape[2],
        )
        self.w_gate_up_flat = gate_up_stacked.reshape(e2i, h)
        self.w_down_t = down_stacked.transpose(1, 2).contiguous()


class MixtralDecodeModel:
    """Decode-only MoE engine; same KV-cache layout as LlamaDecodeModel."""

    def __init__(
        self,
        cfg: MixtralConfig,
        max_batch: int = 256,
        max_seq: int = 2048,
        device: str = "cuda",
        seed: int = 0,
        kv_dtype: str = "bf16",
    ):
        self.cfg = cfg
        self.device = torch.device(device)
        self.dtype = torch.bfloat16
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
        self.layers: List[_MoELayer] = [
            _MoELayer(cfg, self.device, self.dtype, gen)
            for _ in range(cfg.num_layers)
        ]
        self.final_norm = torch.ones(
            cfg.hidden_size, device=self.device, dtype=self.dtype
        )
        self.lm_head = torch.empty(
            cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=self.dtype
        ).normal_(0.0, 0.02, generator=gen)

        self.k_cache = [
            torch.zeros(
                max_batch, cfg.num_kv_heads, max_seq, cfg.head_dim,
                device=self.device, dtype=self.cache_dtype,
            )
            for _ in range(cfg.num_layers)
        ]
        self.v_cache = [
            torch.zeros_like(self.k_cache[0]) for _ in range(cfg.num_layers)
        ]
        self.context_lens = torch.zeros(
            max_batch, dtype=torch.int32, device=self.device
        )
        self.scale = 1.0 / math.sqrt(cfg.head_dim)

    def reset(self, batch: int, context_len: int) -> None:
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

    def _moe_mlp(self, layer: _MoELayer, h2: torch.Tensor) -> torch.Tensor:
        """MoE MLP; dense-batched for decode batches (B ≤ 64).

        Decode MoE on MI355X is weight-streaming-bound: once B ≳
        top_k·E/2 every expert's weights stream from HBM each step, so a
        GEMM over all B tokens costs the same as one over the ~B·k/E
        routed tokens. Running every expert on the full batch as ONE
        strided-batched GEMM pair per layer (torch.bmm → hipBLASLt
        strided-batched) replaces 3·E launches + a per-expert
        `mask.nonzero()` that forces a device→host sync — E·layers = 256
        pipeline stalls per step, the dominant cost of the sparse path
        (measured 48.6 ms at B=64 vs an ~11 ms streaming floor,
        profiles/calibration_mixtral.json). Non-routed expert outputs are
        zero-weighted in the combine einsum. Larger batches take the
        sparse gather path (compute starts to matter), which is also the
        numerics reference for the GPU tests.
        """
        cfg = self.cfg
        B = h2.shape[0]
        router_logits = (h2.float() @ layer.w_router.t().float())  # [B, E]
        weights, selected = torch.topk(router_logits, cfg.top_k, dim=-1)
        weights = torch.softmax(weights, dim=-1).to(h2.dtype)  # [B, K]

        if B <= 64:
            return self._moe_mlp_dense(layer, h2, weights, selected)
        return self._moe_mlp_sparse(layer, h2, weights, selected)

    def _moe_mlp_dense(self, layer, h2, weights, selected):
        cfg = self.cfg
        B = h2.shape[0]
        E, I = cfg.num_experts, cfg.intermediate_size
        w_full = torch.zeros(B, E, device=h2.device, dtype=h2.dtype)
        w_full.scatter_(1, selected, weights)
        # all experts' gate_up as one plain GEMM: [B, H] @ [H, E*2I]
        gate_up = ops.linear(h2, layer.w_gate_up_flat)  # [B, E*2I]
        act = ops.silu_mul_fused(
            gate_up.reshape(B * E, 2 * I)
        ).reshape(B, E, I)
        # down per expert: bmm over contiguous pre-transposed weights
        y = torch.bmm(act.transpose(0, 1).contiguous(), layer.w_down_t)
        # combine: broadcast-mul + sum — NOT einsum, which routes this
        # 4 MB contraction through an 11 ms path on ROCm (measured;
        # profiles/moe_dense_breakdown.txt)
        return (y * w_full.t().unsqueeze(-1)).sum(dim=0)

    def _moe_mlp_sparse(self, layer, h2, weights, selected):
        cfg = self.cfg
        out = torch.zeros_like(h2)
        for e in range(cfg.num_experts):
            # tokens with expert e among their top-k
            mask = selected == e  # [B, K]
            token_idx, k_idx = mask.nonzero(as_tuple=True)
            if token_idx.numel() == 0:
                continue
            x_e = h2.index_select(0, token_idx)
            gate_up = ops.linear(x_e, layer.w_gate_up[e])
            act = ops.silu_mul_fused(gate_up)
            y = ops.linear(act, layer.w_down[e])
            w_e = weights[token_idx, k_idx].unsqueeze(-1)
            out.index_add_(0, token_idx, (y * w_e).to(out.dtype))
        return out

    @torch.no_grad()
    def decode_step(self, token_ids: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        B = token_ids.shape[0]
        positions = self.context_lens[:B].clone()
        ctx = positions + 1  # hoisted out of the layer loop

        x = self.embed.index_select(0, token_ids)
        residual: Optional[torch.Tensor] = None

        for li, layer in enumerate(self.layers):
            if residual is None:
                residual = x.clone()
                h = ops.rmsnorm(x, layer.input_norm, None, cfg.rms_eps)
            else:
                h = ops.rmsnorm(x, layer.input_norm, residual, cfg.rms_eps)

            qkv = ops.linear(h, layer.wqkv)
            q = ops.rope_append_kv(
                qkv, self.k_cache[li][:B], self.v_cache[li][:B],
                positions, cfg.num_q_heads, cfg.num_kv_heads, cfg.rope_theta,
            )
            attn = ops.gqa_decode_attn(
                q, self.k_cache[li][:B], self.v_cache[li][:B], ctx, self.scale
            )
            x = ops.linear(attn.reshape(B, cfg.q_size), layer.wo)

            h2 = ops.rmsnorm(x, layer.post_attn_norm, residual, cfg.rms_eps)
            x = self._moe_mlp(layer, h2)

        final = ops.rmsnorm(x, self.final_norm, residual, cfg.rms_eps)
        logits = ops.linear(final, self.lm_head)
        self.context_lens[:B] += 1
        return logits

    @torch.no_grad()
    def prefill(self, token_ids: torch.Tensor) -> torch.Tensor:
        """Prefill [B, S]: one causal pass (MFMA flash-prefill kernel on
        GPU), fills the KV caches, returns last-position logits.

        The expert MLP takes the SPARSE gather path here: prefill is
        compute-bound (each expert sees ~T·k/E of T = B·S tokens, and
        T is large), so dense-batching would genuinely 4× the MLP FLOPs —
        the opposite trade from decode (see _moe_mlp). The per-expert
        `nonzero()` syncs amortize over one large pass.
        """
        cfg = self.cfg
        B, S = token_ids.shape
        T = B * S
        if S > self.max_seq:
            raise ValueError(f"S={S} exceeds max_seq={self.max_seq}")
        positions = (
            torch.arange(S, device=self.device, dtype=torch.int32).repeat(B)
        )

        x = self.embed.index_select(0, token_ids.reshape(-1))
        residual = None
        causal = None
        if not x.is_cuda:
            causal = torch.full(
                (S, S), float("-inf"), device=self.device, dtype=torch.float32
            ).triu(1)

        for li, layer in enumerate(self.layers):
            if residual is None:
                residual = x.clone()
                h = ops.rmsnorm(x, layer.input_norm, None, cfg.rms_eps)
            else:
                h = ops.rmsnorm(x, layer.input_norm, residual, cfg.rms_eps)

            qkv = h @ layer.wqkv.t()
            q, k, v = qkv.split([cfg.q_size, cfg.kv_size, cfg.kv_size], dim=-1)
            q = q.reshape(T, cfg.num_q_heads, cfg.head_dim).contiguous()
            k = k.reshape(T, cfg.num_kv_heads, cfg.head_dim).contiguous()
            ops.rope(q, k, positions, cfg.rope_theta)
            k_b = k.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
            v_b = v.reshape(B, S, cfg.num_kv_heads, cfg.head_dim)
            self.k_cache[li][:B, :, :S].copy_(k_b.permute(0, 2, 1, 3))
            self.v_cache[li][:B, :, :S].copy_(v_b.permute(0, 2, 1, 3))

            if q.is_cuda:
                attn = ops.prefill_attn(
                    q, self.k_cache[li], self.v_cache[li], B, S, self.scale
                ).reshape(T, cfg.q_size)
            else:
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
                attn = torch.bmm(p, vh)
                attn = (
                    attn.reshape(B, cfg.num_q_heads, S, cfg.head_dim)
                    .permute(0, 2, 1, 3)
                    .reshape(T, cfg.q_size)
                ).contiguous()
            x = attn @ layer.wo.t()

            h2 = ops.rmsnorm(x, layer.post_attn_norm, residual, cfg.rms_eps)
            router_logits = h2.float() @ layer.w_router.t().float()
            weights, selected = torch.topk(router_logits, cfg.top_k, dim=-1)
            weights = torch.softmax(weights, dim=-1).to(h2.dtype)
            x = self._moe_mlp_sparse(layer, h2, weights, selected)

        final = ops.rmsnorm(x, self.final_norm, residual, cfg.rms_eps)
        last = final.reshape(B, S, cfg.hidden_size)[:, -1].contiguous()
        logits = last @ self.lm_head.t()
        self.context_lens.zero_()
        self.context_lens[:B] = S
        return logits
