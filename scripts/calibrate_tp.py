#!/usr/bin/env python3
"""Per-TP-degree calibration over RCCL/xGMI.

Run on one node (torchrun, one rank per GPU); measures ITL(batch) of the
TP-sharded decode model at the launched TP degree, fits α/β, and writes a
capacity record JSON keyed by gpu_count — the measurements the capacity
store and Inferno consume for TP variants:

  torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 8 \
      scripts/calibrate_tp.py --model 70b --batches 1 8 16
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="8b", choices=["8b", "70b", "tiny"])
    p.add_argument("--batches", type=int, nargs="+", default=[1, 8, 32, 64])
    p.add_argument("--context", type=int, default=512)
    p.add_argument("--iters", type=int, default=5)
    p.add_argument("--out", default="gpurun_out/calibration_tp.json")
    args = p.parse_args()

    import torch
    import torch.distributed as dist

    from wva_amd.calibration.itl_benchmark import fit_itl_curve
    from wva_amd.calibration.model import LLAMA_3_8B, LLAMA_3_70B, TINY
    from wva_amd.calibration.tp_model import TPLlamaDecodeModel

    cfg = {"8b": LLAMA_3_8B, "70b": LLAMA_3_70B, "tiny": TINY}[args.model]
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        dist.init_process_group("nccl" if torch.cuda.is_available() else "gloo")
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))

    model = TPLlamaDecodeModel(
        cfg, max_batch=max(args.batches), max_seq=args.context + 64
    )
    itls = []
    for b in args.batches:
        model.reset(b, args.context)
        tokens = torch.randint(0, cfg.vocab_size, (b,), device="cuda")
        for _ in range(3):
            model.decode_step(tokens)
        torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            model.decode_step(tokens)
        torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
        itls.append((time.perf_counter() - t0) * 1000.0 / args.iters)

    alpha, beta, r2 = fit_itl_curve(args.batches, itls)
    if rank == 0:
        record = {
            "model": cfg.name,
            "gpu_count": world,
            "parallelism": f"tp{world}",
            "alpha_ms": alpha,
            "beta_ms": beta,
            "r_squared": r2,
            "batch_sizes": args.batches,
            "itl_ms": itls,
            "context": args.context,
        }
        print(json.dumps(record))
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            json.dump(record, f, indent=2)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
