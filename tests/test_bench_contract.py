"""bench.py contract tests: single-process JSON output and the
2-process gloo (world_size 2) distributed path — the multi-process CPU
coverage for the path torchrun exercises on GPU nodes.
"""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def run_bench(args, env=None):
    e = dict(os.environ)
    if env:
        e.update(env)
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + args,
        capture_output=True, text=True, timeout=420, env=e, cwd=REPO,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    last = out.stdout.strip().splitlines()[-1]
    return json.loads(last)


class TestBenchContract:
    def test_single_process_json(self):
        d = run_bench(["--steps", "6", "--warmup", "2"])
        assert REQUIRED_KEYS <= set(d)
        assert d["unit"] == "%"
        assert d["higher_is_better"] is True
        assert d["scaling"] == "weak"
        assert d["dtype"] == "bf16"
        assert d["data"] == "synthetic"
        assert d["steps"] == 6 and d["warmup"] == 2
        assert 0 <= d["value"] <= 100
        assert d["ms_per_step"] > 0
        assert d["config"]["model"] == "meta-llama/Llama-3.1-8B"

    @pytest.mark.timeout(420)
    def test_two_process_gloo(self):
        out = subprocess.run(
            [
                sys.executable, "-m", "torch.distributed.run",
                "--nnodes=1", "--nproc-per-node", "2",
                "--master-addr", "127.0.0.1", "--master-port", "29531",
                os.path.join(REPO, "bench.py"),
                "--gpus", "2", "--steps", "5", "--warmup", "2",
            ],
            capture_output=True, text=True, timeout=400, cwd=REPO,
        )
        assert out.returncode == 0, out.stderr[-2000:]
        lines = [
            line for line in out.stdout.strip().splitlines()
            if line.startswith("{")
        ]
        assert len(lines) == 1  # only rank 0 prints
        d = json.loads(lines[-1])
        assert d["n_gpus"] == 2
        assert d["config"]["parallelism"] == "dp2"
