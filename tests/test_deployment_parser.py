"""vLLM deployment-args parser tests.

Mirrors reference saturation_v2/deployment_parser_test.go coverage.
"""
from wva_amd.analyzers.deployment_parser import (
    VLLMEngineParams,
    classify_output_length,
    parse_vllm_args,
    split_shell_string,
)
from wva_amd.kube.objects import Container, Deployment, EnvVar, PodTemplateSpec


def deploy_with(command=None, args=None, env=None):
    return Deployment(
        template=PodTemplateSpec(
            containers=[
                Container(
                    command=command or [],
                    args=args or [],
                    env=[EnvVar(name=k, value=v) for k, v in (env or {}).items()],
                )
            ]
        )
    )


class TestParse:
    def test_defaults(self):
        p = parse_vllm_args(None)
        assert p.gpu_memory_utilization == 0.9
        assert p.block_size == 16
        assert p.tensor_parallel_size == 1
        assert p.max_num_seqs == 256
        assert p.is_v1_engine and p.chunked_prefill_enabled
        assert p.effective_max_batched_tokens == 8192  # V1 chunked default

    def test_key_value_equals(self):
        d = deploy_with(args=["--gpu-memory-utilization=0.95", "--block-size=32"])
        p = parse_vllm_args(d)
        assert p.gpu_memory_utilization == 0.95
        assert p.block_size == 32

    def test_key_value_space(self):
        d = deploy_with(args=["--tensor-parallel-size", "8", "--max-num-seqs", "512"])
        p = parse_vllm_args(d)
        assert p.tensor_parallel_size == 8
        assert p.max_num_seqs == 512

    def test_underscore_hyphen_normalization(self):
        d = deploy_with(args=["--max_num_batched_tokens=4096"])
        p = parse_vllm_args(d)
        assert p.max_num_batched_tokens == 4096
        assert p.effective_max_batched_tokens == 4096

    def test_shell_command(self):
        d = deploy_with(
            command=["/bin/sh", "-c"],
            args=["vllm serve meta-llama/Llama-3.1-8B --max-model-len 8192 "
                  "--tensor-parallel-size=4"],
        )
        # collectArgs merges command+args then detects sh -c
        d.template.containers[0].command = [
            "/bin/sh",
            "-c",
            "vllm serve meta-llama/Llama-3.1-8B --max-model-len 8192 "
            "--tensor-parallel-size=4",
        ]
        d.template.containers[0].args = []
        p = parse_vllm_args(d)
        assert p.max_model_len == 8192
        assert p.tensor_parallel_size == 4

    def test_shell_quoting(self):
        toks = split_shell_string('vllm serve "my model" --kv-cache-dtype \'fp8\'')
        assert toks == ["vllm", "serve", "my model", "--kv-cache-dtype", "fp8"]

    def test_boolean_flag(self):
        d = deploy_with(args=["--enforce-eager", "--gpu-memory-utilization", "0.8"])
        p = parse_vllm_args(d)
        assert p.enforce_eager
        assert p.gpu_memory_utilization == 0.8

    def test_vllm_use_v1_zero(self):
        d = deploy_with(args=[], env={"VLLM_USE_V1": "0"})
        p = parse_vllm_args(d)
        assert not p.is_v1_engine
        assert p.effective_max_batched_tokens == 2048  # V0 chunked? no: V0 unchunked
        # V0 with no chunked-prefill flag and no max-model-len → 2048 fallback

    def test_v0_unchunked_long_context(self):
        d = deploy_with(
            args=["--max-model-len", "32768"], env={"VLLM_USE_V1": "0"}
        )
        p = parse_vllm_args(d)
        assert p.effective_max_batched_tokens == 32768

    def test_v0_chunked(self):
        d = deploy_with(
            args=["--enable-chunked-prefill"], env={"VLLM_USE_V1": "0"}
        )
        p = parse_vllm_args(d)
        assert p.effective_max_batched_tokens == 2048

    def test_bad_value_keeps_default(self):
        d = deploy_with(args=["--block-size=NaNsense"])
        p = parse_vllm_args(d)
        assert p.block_size == 16

    def test_num_gpu_blocks_override(self):
        d = deploy_with(args=["--num-gpu-blocks-override=150000"])
        p = parse_vllm_args(d)
        assert p.num_gpu_blocks_override == 150000


class TestCompatibility:
    def test_compatible(self):
        a = VLLMEngineParams(effective_max_batched_tokens=8192)
        b = VLLMEngineParams(effective_max_batched_tokens=8192)
        assert a.is_capacity_compatible(b)

    def test_incompatible_tp(self):
        a = VLLMEngineParams(tensor_parallel_size=8)
        b = VLLMEngineParams(tensor_parallel_size=4)
        assert not a.is_capacity_compatible(b)

    def test_none(self):
        a = VLLMEngineParams()
        assert not a.is_capacity_compatible(None)


class TestClassifyOutputLength:
    def test_buckets(self):
        assert classify_output_length(0) == "short"
        assert classify_output_length(99.9) == "short"
        assert classify_output_length(100) == "medium"
        assert classify_output_length(499) == "medium"
        assert classify_output_length(500) == "long"
        assert classify_output_length(5000) == "long"
