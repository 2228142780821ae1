"""hipGraph-captured decode stepping for MI355X.

A Llama-8B decode step issues ~290 kernel launches (32 layers × (4 HIP
kernels + 3 hipBLASLt GEMMs) + embed/head); at small batch the step is
launch-bound, and launch overhead lands directly in the α of the
ITL(batch) = α + β·batch service model the autoscaler consumes
(SURVEY §5). Capturing one decode step in a hipGraph
(`torch.cuda.CUDAGraph` is hipGraph on ROCm) replays the whole step as
one graph launch.

The engines in this package are graph-safe by construction:
  * all dynamic state (positions, context_lens, KV contents) lives in
    device tensors the kernels read at execution time — no host reads;
  * `gqa_decode_attn` picks its split-KV factor from static shapes only
    (batch, Hk, max_seq), so the captured launch geometry is valid for
    every replay;
  * in-place `context_lens += 1` advances sequence state inside the
    graph, so replay N decodes position ctx+N without re-capture.

Capture requires a GPU; construction raises on CPU-only hosts rather
than silently falling back to eager (the fail-loud rule for the HIP
path).
"""
from __future__ import annotations

import torch


class GraphedDecoder:
    """Wrap a decode engine's `decode_step` in a captured hipGraph.

    The wrapped engine must expose `decode_step(token_ids) -> logits`,
    `context_lens`, `max_seq` and `device`. One GraphedDecoder serves ONE
    batch size (graphs freeze shapes); build one per calibration batch.

    Warm-up runs `warmup_steps` eager steps (hipBLASLt heuristics,
    allocator) and capture itself consumes one more position, so the
    caller must reset the engine state after construction and before
    timed replays — `reset_to` does both.
    """

    def __init__(self, engine, batch: int, warmup_steps: int = 3):
        if not torch.cuda.is_available():
            raise RuntimeError(
                "GraphedDecoder requires a GPU (hipGraph capture)"
            )
        self.engine = engine
        self.batch = batch
        self.static_tokens = torch.zeros(
            batch, dtype=torch.long, device=engine.device
        )
        start_ctx = int(engine.context_lens[:batch].max().item())
        if start_ctx + warmup_steps + 1 >= engine.max_seq:
            raise ValueError(
                f"context {start_ctx} + warmup {warmup_steps} + capture "
                f"would overrun max_seq={engine.max_seq}"
            )
        # warm up on a side stream (standard capture recipe): hipBLASLt
        # solution selection and allocator pools must be settled before
        # capture or their setup work is baked into the graph.
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_steps):
                self.engine.decode_step(self.static_tokens)
        torch.cuda.current_stream().wait_stream(side)

        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_logits = self.engine.decode_step(self.static_tokens)

    def reset_to(self, batch: int, context_len: int) -> None:
        """Re-seed KV/context state; graph replays continue from here."""
        if batch != self.batch:
            raise ValueError(
                f"graph captured for batch={self.batch}, got {batch}"
            )
        self.engine.reset(batch, context_len)

    def decode_step(self, token_ids: torch.Tensor) -> torch.Tensor:
        """One graph replay; returns the static logits buffer.

        The returned tensor is OVERWRITTEN by the next replay — callers
        that keep logits across steps must clone.
        """
        self.static_tokens.copy_(token_ids, non_blocking=True)
        self.graph.replay()
        return self.static_logits
