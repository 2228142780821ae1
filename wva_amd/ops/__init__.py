"""MI355X (gfx950) fused op library.

Hand-written HIP/CDNA4 kernels for the calibration decode path's hot
non-GEMM ops (GEMMs go through hipBLASLt via torch.matmul — guide rule:
library GEMMs stay in the library). The extension is built in-tree
(`wva_amd/ops/_wva_ops*.so`, see setup.py / __graft_entry__.build).

Fail-loud contract: on a ROCm GPU the HIP extension MUST load — there is
no silent eager fallback on the GPU path (the driver verifies the .so is
actually loaded). Pure-CPU processes (unit tests, the control plane) use
the fp32 torch reference implementations below, which are also the
numerics references for the GPU tests.
"""
from __future__ import annotations

import glob
import importlib.util
import math
import os
from typing import Optional

import torch

_HERE = os.path.dirname(os.path.abspath(__file__))
_ext = None
_ext_error: Optional[str] = None


def _try_load_extension():
    global _ext, _ext_error
    if _ext is not None:
        return _ext
    candidates = sorted(glob.glob(os.path.join(_HERE, "_wva_ops*.so")))
    if not candidates:
        _ext_error = (
            f"HIP extension _wva_ops*.so not found in {_HERE}; build it with "
            "`python -m wva_amd.ops.build` (or __graft_entry__.build())"
        )
        return None
    spec = importlib.util.spec_from_file_location("_wva_ops", candidates[0])
    mod = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(mod)
    except Exception as e:  # noqa: BLE001
        _ext_error = f"failed to load {candidates[0]}: {e}"
        return None
    _ext = mod
    return _ext


def extension_available() -> bool:
    return _try_load_extension() is not None


def _require_ext():
    ext = _try_load_extension()
    if ext is None:
        raise RuntimeError(
            f"wva_amd.ops: GPU tensor passed but HIP extension unavailable: "
            f"{_ext_error}"
        )
    return ext


# --- fp32 torch reference implementations (CPU tests + GPU numerics refs) ---


def rmsnorm_ref(
    input: torch.Tensor,
    weight: torch.Tensor,
    residual: Optional[torch.Tensor] = None,
    eps: float = 1e-5,
):
    x = input.float()
    if residual is not None:
        x = x + residual.float()
    folded = x
    rms = torch.rsqrt(x.pow(2).mean(dim=-1, keepdim=True) + eps)
    out = x * rms * weight.float()
    return out, folded


def rope_ref(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
    theta: float = 500000.0,
):
    """NeoX half-rotation RoPE, fp32."""

    def rot(x):
        t, h, d = x.shape
        half = d // 2
        xf = x.float()
        inv_freq = theta ** (
            -2.0 * torch.arange(half, dtype=torch.float32, device=x.device) / d
        )
        angle = positions.float()[:, None] * inv_freq[None, :]  # [T, half]
        cos = angle.cos()[:, None, :]
        sin = angle.sin()[:, None, :]
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)

    return rot(q), rot(k)


def silu_mul_ref(gate: torch.Tensor, up: torch.Tensor):
    g = gate.float()
    return torch.nn.functional.silu(g) * up.float()


def gqa_decode_attn_ref(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
):
    """fp32 reference: per-sequence masked softmax attention.
    Cache layout is head-major [B, Hk, S, D]."""
    B, Hq, D = q.shape
    _, Hk, S, _ = k_cache.shape
    G = Hq // Hk
    out = torch.zeros(B, Hq, D, dtype=torch.float32, device=q.device)
    for b in range(B):
        ctx = int(context_lens[b])
        k = k_cache[b, :, :ctx].float()  # [Hk, ctx, D]
        v = v_cache[b, :, :ctx].float()
        for h in range(Hq):
            kvh = h // G
            scores = (k[kvh] @ (q[b, h].float() * scale))  # [ctx]
            p = torch.softmax(scores, dim=0)
            out[b, h] = p @ v[kvh]
    return out


# --- dispatching public ops (GPU → HIP kernel, CPU → reference) ---


def rmsnorm(
    input: torch.Tensor,
    weight: torch.Tensor,
    residual: Optional[torch.Tensor] = None,
    eps: float = 1e-5,
) -> torch.Tensor:
    if input.is_cuda:
        return _require_ext().rmsnorm(input, weight, residual, eps)
    out, folded = rmsnorm_ref(input, weight, residual, eps)
    if residual is not None:
        residual.copy_(folded.to(residual.dtype))
    return out.to(input.dtype)


def rope(
    q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
    theta: float = 500000.0,
) -> None:
    if q.is_cuda:
        _require_ext().rope(q, k, positions.to(torch.int32), theta)
        return
    qr, kr = rope_ref(q, k, positions, theta)
    q.copy_(qr.to(q.dtype))
    k.copy_(kr.to(k.dtype))


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if gate.is_cuda:
        return _require_ext().silu_mul(gate, up)
    return silu_mul_ref(gate, up).to(gate.dtype)


def rope_append_kv(
    qkv: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    positions: torch.Tensor,
    num_q_heads: int,
    num_kv_heads: int,
    theta: float = 500000.0,
) -> torch.Tensor:
    """Decode fast path: strided qkv row -> RoPE'd q + KV-cache append.
    Returns contiguous q [B, Hq, D]."""
    if qkv.is_cuda:
        return _require_ext().rope_append_kv(
            qkv, k_cache, v_cache, positions.to(torch.int32),
            num_q_heads, num_kv_heads, theta,
        )
    B = qkv.shape[0]
    D = k_cache.shape[3]
    q = qkv[:, : num_q_heads * D].reshape(B, num_q_heads, D).contiguous()
    k = (
        qkv[:, num_q_heads * D : (num_q_heads + num_kv_heads) * D]
        .reshape(B, num_kv_heads, D)
        .contiguous()
    )
    v = qkv[:, (num_q_heads + num_kv_heads) * D :].reshape(B, num_kv_heads, D)
    qr, kr = rope_ref(q, k, positions, theta)
    q.copy_(qr.to(q.dtype))
    k.copy_(kr.to(k.dtype))
    # head-major cache [B, Hk, S, D]
    idx = torch.arange(B)
    k_cache[idx, :, positions.long()] = k.to(k_cache.dtype)
    v_cache[idx, :, positions.long()] = v.to(v_cache.dtype)
    return q


def silu_mul_fused(gate_up: torch.Tensor) -> torch.Tensor:
    """SwiGLU on the fused [rows, 2*inter] gate_up GEMM output."""
    if gate_up.is_cuda:
        return _require_ext().silu_mul_fused(gate_up)
    inter = gate_up.shape[1] // 2
    return silu_mul_ref(gate_up[:, :inter], gate_up[:, inter:]).to(
        gate_up.dtype
    )


def gqa_decode_attn(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    context_lens: torch.Tensor,
    scale: Optional[float] = None,
) -> torch.Tensor:
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _require_ext().gqa_decode_attn(
            q, k_cache, v_cache, context_lens.to(torch.int32), scale
        )
    return gqa_decode_attn_ref(q, k_cache, v_cache, context_lens, scale).to(
        q.dtype
    )


def enable_tuned_gemms() -> bool:
    """Load the committed TunableOp GEMM solution table for gfx950.

    `wva_amd/ops/tunableop_gfx950.csv` holds hipBLASLt/rocBLAS solution
    choices autotuned on MI355X for the decode GEMM shapes (qkv/o/
    gate_up/down/lm_head at batches 1-64) — measured α drops ~10% vs the
    default heuristics (profiles/tunableop_ab.json). Solutions are keyed
    to PyTorch/hipBLASLt versions via the CSV's Validator rows; on a
    mismatched stack TunableOp discards them and falls back to defaults.
    Returns True if the table was loaded.
    """
    import os as _os

    import torch as _torch

    path = _os.path.join(_os.path.dirname(__file__), "tunableop_gfx950.csv")
    if not _torch.cuda.is_available() or not _os.path.exists(path):
        return False
    tun = _torch.cuda.tunable
    tun.enable(True)
    tun.tuning_enable(False)  # read-only: never autotune in production
    tun.read_file(path)
    return True


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Decode linear y = x @ w.T ([M,K] @ [N,K]^T).

    Dispatch is measurement-driven and currently ALWAYS hipBLASLt: the
    in-tree weight-streaming MFMA kernel (csrc/skinny_gemm.hip, exposed
    as ext.skinny_linear) beat hipBLASLt on isolated small-N/small-M
    microbenches but loses IN CONTEXT inside the decode step, where the
    activation matrix is not L2-hot between layers (decode b=1 measured
    5.27 ms/step pure-blaslt vs 5.41 with the kernel dispatched on
    qkv/o). Full history: profiles/skinny_gemm_ab.txt and
    docs/mi355x-kernels.md (negative results). Set WVA_FORCE_SKINNY=1
    to re-enable the measured window for experiments.
    """
    if (
        os.environ.get("WVA_FORCE_SKINNY")
        and x.is_cuda
        and x.shape[0] <= 16
        and x.shape[1] % 128 == 0
        and x.shape[1] <= 8192
        and w.shape[0] <= 8192
    ):
        return _require_ext().skinny_linear(x, w)
    return x @ w.t()


def prefill_attn(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    batch: int,
    seq: int,
    scale: float,
) -> torch.Tensor:
    """Causal GQA flash-prefill over the (populated) head-major caches.

    GPU-only (fail-loud): the CPU prefill reference lives in
    calibration/model.py's matmul branch.
    """
    return _require_ext().prefill_attn(q, k_cache, v_cache, batch, seq, scale)
