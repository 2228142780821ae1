"""Multi-GPU TP tests over RCCL (auto-skip when <2 GPUs).

Pre-staged for the driver's 8-GPU pass:
TP=2 decode numerics against a TP=1 reference, and raw RCCL all-reduce
correctness over xGMI. On a 1-GPU box these skip (not fail), so
`pytest -m gpu` stays green everywhere while the multi-GPU coverage
activates the moment a multi-GPU node appears.

Subprocess-per-rank via torch.distributed.run (one rank per GPU, NCCL
backend = RCCL on ROCm), mirroring how the driver launches bench.py.
"""
import os
import subprocess
import sys
import tempfile

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _gpu_count() -> int:
    try:
        import torch

        return torch.cuda.device_count() if torch.cuda.is_available() else 0
    except Exception:  # noqa: BLE001
        return 0


multi_gpu = pytest.mark.skipif(
    _gpu_count() < 2, reason="needs >=2 ROCm GPUs"
)


def _run_worker(script: str, nproc: int, port: int, timeout: int = 280):
    with tempfile.NamedTemporaryFile("w", suffix=".py", delete=False) as f:
        f.write(script)
        path = f.name
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", str(nproc),
            "--master-addr", "127.0.0.1", "--master-port", str(port),
            path,
        ],
        capture_output=True, text=True, timeout=timeout, env=env,
    )
    os.unlink(path)
    return out


RCCL_ALLREDUCE_WORKER = f'''
import os, torch, torch.distributed as dist
dist.init_process_group("nccl")
rank = dist.get_rank(); world = dist.get_world_size()
torch.cuda.set_device(int(os.environ["LOCAL_RANK"]))
# ring all-reduce correctness at a bucket size representative of the
# TP o-proj partial (hidden 4096, batch 64, bf16)
for numel in (4096, 64 * 4096, 1 << 20):
    x = torch.full((numel,), float(rank + 1), device="cuda",
                   dtype=torch.bfloat16)
    dist.all_reduce(x)
    expect = sum(range(1, world + 1))
    assert torch.all(x == expect), (numel, x[0].item(), expect)
if rank == 0:
    print("RCCL_ALLREDUCE_OK")
dist.destroy_process_group()
'''



TP_ITL_SCALING_WORKER = f'''
import os, sys, time, torch, torch.distributed as dist
sys.path.insert(0, {REPO!r})
from wva_amd.calibration.model import LLAMA_3_8B
from wva_amd.calibration.tp_model import TPLlamaDecodeModel

dist.init_process_group("nccl")
rank = dist.get_rank()
torch.cuda.set_device(int(os.environ["LOCAL_RANK"]))
model = TPLlamaDecodeModel(LLAMA_3_8B, max_batch=8, max_seq=576,
                           device="cuda")
model.reset(8, 512)
tokens = torch.randint(0, LLAMA_3_8B.vocab_size, (8,), device="cuda")
for _ in range(3):
    model.decode_step(tokens)
torch.cuda.synchronize(); dist.barrier()
t0 = time.perf_counter()
for _ in range(8):
    model.decode_step(tokens)
torch.cuda.synchronize(); dist.barrier()
itl = (time.perf_counter() - t0) * 1000.0 / 8
if rank == 0:
    # sanity: a TP=2 8B decode step on MI355X must be < 100 ms and > 0.5 ms
    assert 0.5 < itl < 100.0, itl
    print(f"TP_ITL_OK itl_ms={{itl:.3f}}")
dist.destroy_process_group()
'''


@pytest.mark.gpu
@multi_gpu
@pytest.mark.timeout(300)
def test_rccl_allreduce_correctness():
    out = _run_worker(RCCL_ALLREDUCE_WORKER, nproc=2, port=29611)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "RCCL_ALLREDUCE_OK" in out.stdout


@pytest.mark.gpu
@multi_gpu
@pytest.mark.timeout(300)
def test_tp2_decode_matches_tp1():
    env = dict(os.environ,
               WVA_TP_DEVICE="cuda", WVA_TP_BACKEND="nccl",
               HSA_ENABLE_IPC_MODE_LEGACY="0")
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29612",
            os.path.join(REPO, "tests", "tp_numerics_worker.py"),
        ],
        capture_output=True, text=True, timeout=280, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "TP2_NUMERICS_OK" in out.stdout


@pytest.mark.gpu
@multi_gpu
@pytest.mark.timeout(300)
def test_tp2_8b_itl_sane():
    out = _run_worker(TP_ITL_SCALING_WORKER, nproc=2, port=29613)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "TP_ITL_OK" in out.stdout
