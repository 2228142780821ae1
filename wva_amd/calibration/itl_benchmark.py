"""ITL-vs-batch calibration on MI355X.

Measures the decode iteration time of a LlamaDecodeModel at a sweep of
batch sizes, fits ITL(batch) = α + β·batch by least squares (the Inferno
ServiceParms linear model, reference docs/design/modeling-optimization.md
:52), derives the KV-token capacity from the GPU's free HBM, and emits a
ServiceProfile for the emulator plus a JSON record for profiles/.

This is the reference's offline "parameter estimation" methodology
(docs/tutorials/parameter-estimation.md) turned into an in-repo measured
path on the actual target hardware.
"""
from __future__ import annotations

import json
import time
from dataclasses import asdict, dataclass
from typing import List, Optional, Tuple

import torch

from ..emulator.vllm_sim import ServiceProfile
from .model import LlamaConfig, LlamaDecodeModel


@dataclass
class CalibrationResult:
    model: str
    device: str
    gpu_count: int
    alpha_ms: float
    beta_ms: float
    batch_sizes: List[int]
    itl_ms: List[float]
    r_squared: float
    kv_capacity_tokens: int
    num_gpu_blocks: int
    block_size: int
    decode_tokens_per_s_peak: float
    prefill_tokens_per_s: float = 0.0
    # GPU-busy marker phase (driver-verifiable evidence that the
    # calibration really ran on the device: long enough for SMI sampling)
    busy_marker_s: float = 0.0
    busy_tokens: int = 0

    def to_json(self) -> str:
        return json.dumps(asdict(self), indent=2)


def fit_itl_curve(
    batch_sizes: List[int], itl_ms: List[float]
) -> Tuple[float, float, float]:
    """Least-squares fit ITL = alpha + beta * batch; returns (a, b, R²)."""
    import numpy as np

    x = np.asarray(batch_sizes, dtype=np.float64)
    y = np.asarray(itl_ms, dtype=np.float64)
    A = np.stack([np.ones_like(x), x], axis=1)
    coef, *_ = np.linalg.lstsq(A, y, rcond=None)
    alpha, beta = float(coef[0]), float(coef[1])
    pred = alpha + beta * x
    ss_res = float(((y - pred) ** 2).sum())
    ss_tot = float(((y - y.mean()) ** 2).sum())
    r2 = 1.0 - ss_res / ss_tot if ss_tot > 0 else 1.0
    return alpha, beta, r2


def measure_itl(
    model: LlamaDecodeModel,
    batch: int,
    context_len: int = 512,
    iters: int = 8,
    warmup: int = 3,
    use_graph: bool = False,
) -> float:
    """Median decode-iteration wall time (ms) at the given batch size.

    With `use_graph` the step is replayed as one captured hipGraph
    (calibration/graph.py) — measuring the launch-overhead-free serving
    configuration, which is what a production engine would run.
    """
    model.reset(batch, context_len)
    tokens = torch.randint(
        0, model.cfg.vocab_size, (batch,), device=model.device
    )
    if use_graph:
        from .graph import GraphedDecoder

        dec = GraphedDecoder(model, batch, warmup_steps=max(warmup, 1))
        dec.reset_to(batch, context_len)
        step = dec.decode_step
        step(tokens)  # one replay to settle
    else:
        step = model.decode_step
        for _ in range(warmup):
            step(tokens)
    torch.cuda.synchronize()
    times = []
    for _ in range(iters):
        t0 = time.perf_counter()
        step(tokens)
        torch.cuda.synchronize()
        times.append((time.perf_counter() - t0) * 1000.0)
    times.sort()
    return times[len(times) // 2]


def measure_prefill_tps(
    model: LlamaDecodeModel,
    batch: int = 2,
    seq: int = 1024,
    iters: int = 3,
) -> float:
    """Prefill throughput (prompt tokens/s) — the TTFT model's input
    (ServiceProfile.prefill_tokens_per_s), measured instead of assumed."""
    seq = min(seq, model.max_seq)
    batch = min(batch, model.max_batch)
    tokens = torch.randint(
        0, model.cfg.vocab_size, (batch, seq), device=model.device
    )
    model.prefill(tokens)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        model.prefill(tokens)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return batch * seq * iters / dt


def derive_kv_capacity(
    cfg: LlamaConfig,
    gpu_memory_utilization: float = 0.9,
    block_size: int = 16,
    hbm_bytes: Optional[int] = None,
    kv_dtype_bytes: int = 2,
) -> Tuple[int, int]:
    """(num_gpu_blocks, kv_capacity_tokens) from free HBM after weights —
    the vllm:cache_config_info analog, sized for 288 GB HBM3E. fp8 KV
    (kv_dtype_bytes=1) doubles the token capacity."""
    if hbm_bytes is None:
        hbm_bytes = torch.cuda.get_device_properties(0).total_memory
    budget = int(hbm_bytes * gpu_memory_utilization) - cfg.weight_bytes()
    per_token = cfg.kv_bytes_per_token(kv_dtype_bytes)
    tokens = max(budget // per_token, 0)
    blocks = tokens // block_size
    return blocks, blocks * block_size


def calibrate_service_profile(
    cfg: LlamaConfig,
    batch_sizes: Optional[List[int]] = None,
    context_len: int = 512,
    max_seq: int = 1024,
    gpu_count: int = 1,
    iters: int = 8,
    use_graph: bool = False,
    kv_dtype: str = "bf16",
    busy_seconds: float = 0.0,
) -> Tuple[ServiceProfile, CalibrationResult]:
    """Measure on the current GPU and return an emulator ServiceProfile
    + the raw calibration record. kv_dtype="fp8" measures the e4m3
    KV-cache serving configuration (decode only; prefill throughput is
    then measured on a separate bf16-cache engine)."""
    if batch_sizes is None:
        batch_sizes = [1, 2, 4, 8, 16, 32, 64]
    max_batch = max(batch_sizes)
    model = LlamaDecodeModel(
        cfg, max_batch=max_batch, max_seq=max_seq, kv_dtype=kv_dtype
    )
    if kv_dtype == "bf16":
        prefill_tps = measure_prefill_tps(model)
    else:
        pf_model = LlamaDecodeModel(cfg, max_batch=2, max_seq=max_seq)
        prefill_tps = measure_prefill_tps(pf_model)
        del pf_model
        torch.cuda.empty_cache()
    itl: List[float] = []
    for b in batch_sizes:
        itl.append(
            measure_itl(model, b, context_len, iters=iters, use_graph=use_graph)
        )
    alpha, beta, r2 = fit_itl_curve(batch_sizes, itl)
    blocks, kv_tokens = derive_kv_capacity(
        cfg, kv_dtype_bytes=1 if kv_dtype == "fp8" else 2
    )
    peak_tps = max(
        b / (t / 1000.0) for b, t in zip(batch_sizes, itl)
    )
    # GPU-busy marker: keep the device saturated with real decode steps
    # for a fixed wall-clock window so out-of-process SMI sampling (the
    # bench driver's gpu_busy probe) cannot miss the GPU phase
    busy_tokens = 0
    busy_elapsed = 0.0
    if busy_seconds > 0:
        b = max_batch
        model.reset(b, context_len)
        toks = torch.randint(0, cfg.vocab_size, (b,), device=model.device)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        while time.perf_counter() - t0 < busy_seconds:
            for _ in range(8):
                model.decode_step(toks)
                busy_tokens += b
            torch.cuda.synchronize()
            if model.context_lens[0].item() >= model.max_seq - 8:
                model.reset(b, context_len)
        busy_elapsed = time.perf_counter() - t0
    result = CalibrationResult(
        model=cfg.name,
        device=torch.cuda.get_device_name(0),
        gpu_count=gpu_count,
        alpha_ms=alpha,
        beta_ms=beta,
        batch_sizes=batch_sizes,
        itl_ms=itl,
        r_squared=r2,
        kv_capacity_tokens=kv_tokens,
        num_gpu_blocks=blocks,
        block_size=16,
        decode_tokens_per_s_peak=peak_tps,
        prefill_tokens_per_s=prefill_tps,
        busy_marker_s=busy_elapsed,
        busy_tokens=busy_tokens,
    )
    profile = ServiceProfile(
        alpha_ms=max(alpha, 0.1),
        beta_ms=max(beta, 0.0),
        max_num_seqs=256,
        num_gpu_blocks=blocks,
        block_size=16,
        prefill_tokens_per_s=prefill_tps,
    )
    # free the model before returning (bench reuses the GPU)
    del model
    torch.cuda.empty_cache()
    return profile, result


def fit_itl_surface(points) -> Tuple[float, float, float, float]:
    """Fit the reference's FULL 3-parameter service model from a
    measured (batch, context, itl_ms) surface.

    The Inferno iteration-time model (queueanalyzer.go:261-279) is
      iterTime = α + n·(β·tokensCompute + γ·tokensMemory)
    with tokensMemory ≈ the per-request KV footprint (context tokens).
    A fixed-context calibration can only produce α and a β that silently
    absorbs γ·ctx at that one context; long-context serving then looks
    as cheap as short. This fits the separable surface
      ITL(n, ctx) = α + n·(β_eff + γ·ctx)
    by least squares over the measured grid (β_eff = β·tokensCompute is
    converted back by ServiceParmsSpec.from_itl_surface at the request
    mix). Returns (alpha_ms, beta_eff_ms, gamma_ms_per_ctx_token, R²).
    """
    import numpy as np

    pts = list(points)
    A = np.array([[1.0, b, b * c] for b, c, _ in pts], dtype=np.float64)
    y = np.array([t for _, _, t in pts], dtype=np.float64)
    coef, *_ = np.linalg.lstsq(A, y, rcond=None)
    alpha, beta_eff, gamma = (float(v) for v in coef)
    if beta_eff < 0.0 or gamma < 0.0:
        # physical constraint: service time is non-decreasing in batch
        # and context (a weight-streaming-dominated model like 70B can
        # fit a slightly negative β that would make the queueing model's
        # service rates non-monotone) — refit with the offending term
        # dropped
        cols = [0] + ([1] if beta_eff >= 0.0 else []) \
            + ([2] if gamma >= 0.0 else [])
        coef_c, *_ = np.linalg.lstsq(A[:, cols], y, rcond=None)
        full = [0.0, 0.0, 0.0]
        for idx, c_ in zip(cols, coef_c):
            full[idx] = float(c_)
        alpha, beta_eff, gamma = full
        coef = np.array(full)
    pred = A @ coef
    ss_res = float(((y - pred) ** 2).sum())
    ss_tot = float(((y - y.mean()) ** 2).sum())
    r2 = 1.0 - ss_res / ss_tot if ss_tot > 0 else 1.0
    return alpha, max(beta_eff, 0.0), max(gamma, 0.0), r2
