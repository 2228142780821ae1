"""Tensor-parallel Llama decode engine over RCCL/xGMI.

The autoscaler's capacity model is TP-aware: service-rate records are
keyed by (model, accelerator, gpuCount) and the α of the ITL model grows
with the per-layer all-reduce term on xGMI. This engine
measures that directly: Megatron-style TP sharding of the calibration
decode model with `torch.distributed` all-reduce (backend "nccl" IS RCCL
on ROCm; "gloo" for CPU tests).

Sharding (standard column/row parallel):
  * wqkv, w_gate_up: column-parallel — each rank owns Hq/N query heads,
    Hk/N kv heads and I/N intermediate columns; no communication in.
  * wo, w_down: row-parallel — partial sums all-reduced (2 all-reduces
    per layer, ring over 7×153 GB/s xGMI links intra-node).
  * Attention + KV cache: local to the rank's heads.
  * Embedding/lm_head replicated (calibration-scale simplification).

Run per-TP calibration on an 8-GPU node with:
  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node N \
      scripts/calibrate_tp.py --model 8b
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.distributed as dist

from .. import ops
from .model import LlamaConfig


class _TPLayer:
    def __init__(self, cfg: LlamaConfig, shard: "TPShardInfo", device, dtype, gen):
        h, std = cfg.hidden_size, 0.02

        def w(rows, cols):
            return torch.empty(rows, cols, device=device, dtype=dtype).normal_(
                0.0, std, generator=gen
            )

        self.input_norm = torch.ones(h, device=device, dtype=dtype)
        self.post_attn_norm = torch.ones(h, device=device, dtype=dtype)
        # column-parallel qkv: this rank's heads only
        self.wqkv = w(shard.q_size + 2 * shard.kv_size, h)
        # row-parallel o: [H, q_size/N]
        self.wo = w(h, shard.q_size)
        # column-parallel gate_up: [2*I/N, H]
        self.w_gate_up = w(2 * shard.inter, h)
        # row-parallel down: [H, I/N]
        self.w_down = w(h, shard.inter)


class TPShardInfo:
    def __init__(self, cfg: LlamaConfig, tp: int, rank: int):
        if cfg.num_q_heads % tp or cfg.num_kv_heads % tp:
            raise ValueError(
                f"heads ({cfg.num_q_heads}/{cfg.num_kv_heads}) must divide "
                f"by tp={tp}"
            )
        if cfg.intermediate_size % tp:
            raise ValueError("intermediate_size must divide by tp")
        self.tp = tp
        self.rank = rank
        self.num_q_heads = cfg.num_q_heads // tp
        self.num_kv_heads = cfg.num_kv_heads // tp
        self.inter = cfg.intermediate_size // tp
        self.q_size = self.num_q_heads * cfg.head_dim
        self.kv_size = self.num_kv_heads * cfg.head_dim


class TPLlamaDecodeModel:
    """One rank of a TP=N decode engine; construct once per process."""

    def __init__(
        self,
        cfg: LlamaConfig,
        max_batch: int = 64,
        max_seq: int = 1024,
        device: str = "cuda",
        seed: int = 0,
        group: Optional[dist.ProcessGroup] = None,
        kv_dtype: str = "bf16",
    ):
        self.cfg = cfg
        self.kv_dtype = kv_dtype
        self.cache_dtype = (
            torch.float8_e4m3fn if kv_dtype == "fp8" else torch.bfloat16
        )
        self.group = group
        self.tp = dist.get_world_size(group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.shard = TPShardInfo(cfg, self.tp, self.rank)
        self.device = torch.device(device)
        self.dtype = torch.bfloat16
        self.max_batch = max_batch
        self.max_seq = max_seq
        gen = torch.Generator(device=self.device)
        # same seed on every rank for the replicated parts; shard-unique
        # seed for the sharded weights
        gen.manual_seed(seed)
        self.embed = torch.empty(
            cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=self.dtype
        ).normal_(0.0, 0.02, generator=gen)
        gen.manual_seed(seed * 1000 + self.rank + 1)
        self.layers: List[_TPLayer] = [
            _TPLayer(cfg, self.shard, self.device, self.dtype, gen)
            for _ in range(cfg.num_layers)
        ]
        self.final_norm = torch.ones(
            cfg.hidden_size, device=self.device, dtype=self.dtype
        )
        gen.manual_seed(seed)
        self.lm_head = torch.empty(
            cfg.vocab_size, cfg.hidden_size, device=self.device, dtype=self.dtype
        ).normal_(0.0, 0.02, generator=gen)

        self.k_cache = [
            torch.zeros(
                max_batch, self.shard.num_kv_heads, max_seq, cfg.head_dim,
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

    def _all_reduce(self, x: torch.Tensor) -> torch.Tensor:
        if self.tp > 1:
            dist.all_reduce(x, group=self.group)
        return x

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

    @torch.no_grad()
    def decode_step(self, token_ids: torch.Tensor) -> torch.Tensor:
        cfg, shard = self.cfg, self.shard
        B = token_ids.shape[0]
        positions = self.context_lens[:B].clone()

        ctx = positions + 1  # includes the new token (hoisted: one
        # launch per step, not per layer — profiling showed 32 redundant
        # int-add launches/step, profiles/decode8b_r2_kernel_stats.txt)
        x = self.embed.index_select(0, token_ids)
        residual: Optional[torch.Tensor] = None

        for li, layer in enumerate(self.layers):
            if residual is None:
                residual = x.clone()
                h = ops.rmsnorm(x, layer.input_norm, None, cfg.rms_eps)
            else:
                h = ops.rmsnorm(x, layer.input_norm, residual, cfg.rms_eps)

            qkv = ops.linear(h, layer.wqkv)  # local heads only
            q = ops.rope_append_kv(
                qkv, self.k_cache[li][:B], self.v_cache[li][:B],
                positions, shard.num_q_heads, shard.num_kv_heads,
                cfg.rope_theta,
            )
            attn = ops.gqa_decode_attn(
                q, self.k_cache[li][:B], self.v_cache[li][:B], ctx, self.scale
            )
            # row-parallel o-proj: partials all-reduced over RCCL/xGMI
            x = self._all_reduce(ops.linear(attn.reshape(B, shard.q_size), layer.wo))

            h2 = ops.rmsnorm(x, layer.post_attn_norm, residual, cfg.rms_eps)
            gate_up = ops.linear(h2, layer.w_gate_up)
            act = ops.silu_mul_fused(gate_up)
            x = self._all_reduce(ops.linear(act, layer.w_down))

        final = ops.rmsnorm(x, self.final_norm, residual, cfg.rms_eps)
        logits = ops.linear(final, self.lm_head)
        self.context_lens[:B] += 1
        return logits
