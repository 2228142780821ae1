"""vLLM deployment-argument parser.

Parity: reference internal/engines/analyzers/saturation_v2/deployment_parser.go
:13-268 — same flag set, `--a-b`/`--a_b` normalization, `sh -c` shell-string
splitting with quote handling, VLLM_USE_V1 env detection, and the
EffectiveMaxBatchedTokens resolution chain (explicit → 8192 V1-chunked →
2048 V0-chunked → max(maxModelLen, 2048) unchunked → 2048).

vLLM-ROCm note: the flags accepted by vLLM on ROCm are the same CLI surface;
ROCm-specific env (VLLM_ROCM_USE_AITER etc.) does not change the capacity
math, so only capacity-relevant flags are parsed — matching the reference.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from ..kube.objects import Deployment

# Output-length buckets for k2 history keying (constants.go:30-36)
SHORT_OUTPUT_THRESHOLD = 100
MEDIUM_OUTPUT_THRESHOLD = 500


def classify_output_length(avg_output_tokens: float) -> str:
    if avg_output_tokens < SHORT_OUTPUT_THRESHOLD:
        return "short"
    if avg_output_tokens < MEDIUM_OUTPUT_THRESHOLD:
        return "medium"
    return "long"


@dataclass
class VLLMEngineParams:
    gpu_memory_utilization: float = 0.9
    block_size: int = 16
    kv_cache_dtype: str = "auto"
    tensor_parallel_size: int = 1
    num_gpu_blocks_override: int = 0
    max_num_batched_tokens: int = 0
    max_num_seqs: int = 256
    max_model_len: int = 0
    enforce_eager: bool = False
    is_v1_engine: bool = True
    chunked_prefill_enabled: bool = True
    effective_max_batched_tokens: int = 0

    def is_capacity_compatible(self, other: Optional["VLLMEngineParams"]) -> bool:
        if other is None:
            return False
        return (
            self.gpu_memory_utilization == other.gpu_memory_utilization
            and self.block_size == other.block_size
            and self.kv_cache_dtype == other.kv_cache_dtype
            and self.tensor_parallel_size == other.tensor_parallel_size
            and self.num_gpu_blocks_override == other.num_gpu_blocks_override
            and self.effective_max_batched_tokens == other.effective_max_batched_tokens
        )


def split_shell_string(s: str) -> List[str]:
    """Basic shell-like splitting with single/double quote support
    (no expansion — same scope as the reference). All unquoted
    whitespace separates tokens — `sh -c` commands in YAML manifests
    are typically `|` block scalars with newlines — and a bare `\\`
    line continuation is dropped."""
    tokens: List[str] = []
    current: List[str] = []
    in_single = False
    in_double = False
    for ch in s:
        if ch == "'" and not in_double:
            in_single = not in_single
        elif ch == '"' and not in_single:
            in_double = not in_double
        elif ch in " \t\n\r" and not in_single and not in_double:
            if current:
                tokens.append("".join(current))
                current = []
        else:
            current.append(ch)
    if current:
        tokens.append("".join(current))
    return [t for t in tokens if t != "\\"]


def _collect_args(command: List[str], args: List[str]) -> List[str]:
    all_args = list(command) + list(args)
    for i in range(len(all_args) - 1):
        base = all_args[i]
        if (
            base in ("/bin/sh", "/bin/bash", "sh", "bash")
            and all_args[i + 1] == "-c"
            and i + 2 < len(all_args)
        ):
            return split_shell_string(all_args[i + 2])
    return all_args


def _normalize_key(key: str) -> str:
    return key.lstrip("-").replace("-", "_")


def _apply_param(key: str, value: str, params: VLLMEngineParams) -> None:
    try:
        if key == "gpu_memory_utilization":
            params.gpu_memory_utilization = float(value)
        elif key == "block_size":
            params.block_size = int(value)
        elif key == "kv_cache_dtype":
            params.kv_cache_dtype = value
        elif key == "tensor_parallel_size":
            params.tensor_parallel_size = int(value)
        elif key == "num_gpu_blocks_override":
            params.num_gpu_blocks_override = int(value)
        elif key == "max_num_batched_tokens":
            params.max_num_batched_tokens = int(value)
        elif key == "max_num_seqs":
            params.max_num_seqs = int(value)
        elif key == "max_model_len":
            params.max_model_len = int(value)
        elif key == "enforce_eager":
            params.enforce_eager = True
        elif key == "enable_chunked_prefill":
            params.chunked_prefill_enabled = True
    except (TypeError, ValueError):
        # Parse errors preserve the default — graceful degradation since
        # deployment args are operator-controlled.
        pass


def _parse_args(args: List[str], params: VLLMEngineParams) -> None:
    i = 0
    while i < len(args):
        arg = args[i]
        if not arg.startswith("--"):
            i += 1
            continue
        if "=" in arg:
            key, _, value = arg.partition("=")
            key = _normalize_key(key)
        else:
            key = _normalize_key(arg)
            value = ""
            if i + 1 < len(args) and not args[i + 1].startswith("--"):
                value = args[i + 1]
                i += 1
        _apply_param(key, value, params)
        i += 1


def resolve_effective_max_batched_tokens(params: VLLMEngineParams) -> None:
    if params.max_num_batched_tokens > 0:
        params.effective_max_batched_tokens = params.max_num_batched_tokens
        return
    if params.chunked_prefill_enabled:
        params.effective_max_batched_tokens = 8192 if params.is_v1_engine else 2048
        return
    if params.max_model_len > 2048:
        params.effective_max_batched_tokens = params.max_model_len
        return
    params.effective_max_batched_tokens = 2048


def parse_vllm_args(deploy: Optional[Deployment]) -> VLLMEngineParams:
    """Parse vLLM CLI args + env from a Deployment's pod template."""
    params = VLLMEngineParams()
    if deploy is None or not deploy.template.containers:
        resolve_effective_max_batched_tokens(params)
        return params

    for container in deploy.template.containers:
        for env in container.env:
            if env.name == "VLLM_USE_V1" and env.value == "0":
                params.is_v1_engine = False
                params.chunked_prefill_enabled = False
        all_args = _collect_args(container.command, container.args)
        _parse_args(all_args, params)

    if params.is_v1_engine:
        params.chunked_prefill_enabled = True
    resolve_effective_max_batched_tokens(params)
    return params
