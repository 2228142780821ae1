#!/usr/bin/env python3
"""Decode-step workload for rocprofv3 profiling on MI355X.

Runs Llama-3-8B decode iterations at a fixed batch so the kernel trace
captures the steady-state mix (HIP kernels from wva_amd.ops + hipBLASLt
GEMMs). Keep iterations small — rocprof multiplies overhead.

    cd /tmp && export TMPDIR=/tmp
    rocprofv3 --kernel-trace --stats -d gpurun_out/prof -- \
        python scripts/profile_decode.py --batch 64 --iters 10
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=64)
    p.add_argument("--iters", type=int, default=10)
    p.add_argument("--context", type=int, default=512)
    p.add_argument("--model", default="8b", choices=["8b", "tiny"])
    args = p.parse_args()

    import torch
    from wva_amd.calibration.model import LLAMA_3_8B, TINY, LlamaDecodeModel
    from wva_amd.ops import enable_tuned_gemms

    # profile the SERVING configuration: committed TunableOp solutions
    # (bench.py enables the same table before calibration)
    enable_tuned_gemms()
    cfg = LLAMA_3_8B if args.model == "8b" else TINY
    model = LlamaDecodeModel(cfg, max_batch=args.batch, max_seq=args.context + args.iters + 8)
    model.reset(args.batch, args.context)
    tokens = torch.randint(0, cfg.vocab_size, (args.batch,), device="cuda")

    # warmup
    for _ in range(3):
        model.decode_step(tokens)
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(args.iters):
        model.decode_step(tokens)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    itl_ms = dt * 1000.0 / args.iters
    print(
        f"decode: batch={args.batch} ctx={args.context} iters={args.iters} "
        f"ITL={itl_ms:.3f} ms  decode_tps={args.batch / (itl_ms / 1000.0):.0f}"
    )


if __name__ == "__main__":
    main()


def attn_bench():
    """Standalone attention microbench: bytes-based bandwidth estimate."""
    import torch
    from wva_amd import ops

    for batch, hq, hk, ctx in [(64, 32, 8, 512), (64, 32, 8, 2048),
                               (256, 32, 8, 512), (8, 64, 8, 4096)]:
        S = ctx
        q = torch.randn(batch, hq, 128, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(batch, hk, S, 128, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(batch, hk, S, 128, device="cuda", dtype=torch.bfloat16)
        lens = torch.full((batch,), ctx, device="cuda", dtype=torch.int32)
        from wva_amd.ops import _require_ext
        ext = _require_ext()
        scale = 128 ** -0.5
        import time as _t
        for name, fn in [("v3", ext.gqa_decode_attn_v3),
                         ("v4", ext.gqa_decode_attn_v4),
                         ("v5", ext.gqa_decode_attn_v5)]:
            for _ in range(3):
                fn(q, k, v, lens, scale)
            torch.cuda.synchronize()
            n = 20
            t0 = _t.perf_counter()
            for _ in range(n):
                fn(q, k, v, lens, scale)
            torch.cuda.synchronize()
            us = (_t.perf_counter() - t0) / n * 1e6
            kv_bytes = 2 * batch * ctx * hk * 128 * 2
            print(f"attn[{name}] b={batch} hq={hq} hk={hk} ctx={ctx}: "
                  f"{us:.1f} us, KV {kv_bytes/1e6:.1f} MB, "
                  f"{kv_bytes/us/1e3:.2f} TB/s")


def calibrate_70b():
    """Llama-3.1-70B on ONE MI355X: 140 GB bf16 weights fit in 288 GB HBM3E
    (impossible on 192 GB MI300X without TP) — BASELINE config #5 evidence."""
    import json
    import torch
    from wva_amd.calibration.itl_benchmark import calibrate_service_profile
    from wva_amd.calibration.model import LLAMA_3_70B

    profile, result = calibrate_service_profile(
        LLAMA_3_70B, batch_sizes=[1, 4, 8, 16], context_len=256,
        max_seq=512, iters=4,
    )
    print("llama-70b calibration:", result.to_json())
    free, total = torch.cuda.mem_get_info()
    print(f"hbm total={total/2**30:.0f}GiB")
    with open("gpurun_out/calibration_70b.json", "w") as f:
        f.write(result.to_json())


def calibrate_mixtral():
    """Mixtral-8x7B (MoE) on ONE MI355X: ~94 GB bf16 weights resident
    (BASELINE config #4's second model family)."""
    import json
    import time as _t

    import torch

    from wva_amd.calibration.itl_benchmark import fit_itl_curve
    from wva_amd.calibration.moe_model import MIXTRAL_8X7B, MixtralDecodeModel

    batches = [1, 8, 32, 64]
    model = MixtralDecodeModel(MIXTRAL_8X7B, max_batch=max(batches), max_seq=1024)
    itls = []
    for b in batches:
        model.reset(b, 512)
        tokens = torch.randint(0, MIXTRAL_8X7B.vocab_size, (b,), device="cuda")
        for _ in range(3):
            model.decode_step(tokens)
        torch.cuda.synchronize()
        t0 = _t.perf_counter()
        for _ in range(5):
            model.decode_step(tokens)
        torch.cuda.synchronize()
        itls.append((_t.perf_counter() - t0) * 1000.0 / 5)
    alpha, beta, r2 = fit_itl_curve(batches, itls)
    rec = {
        "model": MIXTRAL_8X7B.name,
        "gpu_count": 1,
        "alpha_ms": alpha,
        "beta_ms": beta,
        "r_squared": r2,
        "batch_sizes": batches,
        "itl_ms": itls,
        "weights_gib": MIXTRAL_8X7B.weight_bytes() / 2**30,
        "decode_tokens_per_s_peak": max(
            b / (t / 1000.0) for b, t in zip(batches, itls)
        ),
    }
    print("mixtral calibration:", json.dumps(rec, indent=2))
    import os

    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/calibration_mixtral.json", "w") as f:
        json.dump(rec, f, indent=2)


def graph_ab():
    """Eager vs hipGraph-captured decode A/B on Llama-3-8B: the launch
    overhead shows up in alpha of ITL = alpha + beta*batch."""
    import json

    from wva_amd.calibration.itl_benchmark import fit_itl_curve, measure_itl
    from wva_amd.calibration.model import LLAMA_3_8B, LlamaDecodeModel

    batches = [1, 8, 32, 64]
    model = LlamaDecodeModel(LLAMA_3_8B, max_batch=max(batches), max_seq=1024)
    rec = {}
    for mode, use_graph in [("eager", False), ("graph", True)]:
        itls = [
            measure_itl(model, b, 512, iters=8, use_graph=use_graph)
            for b in batches
        ]
        alpha, beta, r2 = fit_itl_curve(batches, itls)
        rec[mode] = {
            "itl_ms": itls, "alpha_ms": alpha, "beta_ms": beta, "r2": r2,
            "peak_tps": max(b / (t / 1000.0) for b, t in zip(batches, itls)),
        }
        print(f"{mode}: itl={['%.3f' % t for t in itls]} "
              f"alpha={alpha:.3f} beta={beta:.4f} peak={rec[mode]['peak_tps']:.0f}")
    import os
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/graph_ab.json", "w") as f:
        json.dump(rec, f, indent=2)


def tunableop_ab():
    """TunableOp A/B: autotune hipBLASLt GEMM solutions for the decode
    shapes, persist the tuning table, measure eager ITL before/after."""
    import json
    import os

    import torch

    from wva_amd.calibration.itl_benchmark import fit_itl_curve, measure_itl
    from wva_amd.calibration.model import LLAMA_3_8B, LlamaDecodeModel

    os.makedirs("gpurun_out", exist_ok=True)
    batches = [1, 8, 32, 64]
    model = LlamaDecodeModel(LLAMA_3_8B, max_batch=max(batches), max_seq=1024)

    base = [measure_itl(model, b, 512, iters=8) for b in batches]
    a0, b0, _ = fit_itl_curve(batches, base)

    tun = torch.cuda.tunable
    tun.enable(True)
    tun.tuning_enable(True)
    tun.set_filename("gpurun_out/tunableop_8b.csv")
    for b in batches:  # tuning happens on first call per GEMM shape
        measure_itl(model, b, 512, iters=2)
    tun.write_file() if hasattr(tun, "write_file") else None
    tun.tuning_enable(False)

    tuned = [measure_itl(model, b, 512, iters=8) for b in batches]
    a1, b1, _ = fit_itl_curve(batches, tuned)
    rec = {
        "baseline": {"itl_ms": base, "alpha_ms": a0, "beta_ms": b0},
        "tunableop": {"itl_ms": tuned, "alpha_ms": a1, "beta_ms": b1},
    }
    print("tunableop A/B:", json.dumps(rec, indent=2))
    with open("gpurun_out/tunableop_ab.json", "w") as f:
        json.dump(rec, f, indent=2)


def skinny_gemm_bench():
    """skinny_linear vs hipBLASLt (torch.matmul) on the decode shapes."""
    import time as _t

    import torch

    from wva_amd.ops import _require_ext
    ext = _require_ext()

    shapes = [  # (label, M, N, K)
        ("qkv", 64, 6144, 4096), ("o", 64, 4096, 4096),
        ("gate_up", 64, 28672, 4096), ("down", 64, 4096, 14336),
        ("lm_head", 64, 128256, 4096),
        ("qkv_b1", 1, 6144, 4096), ("gate_up_b1", 1, 28672, 4096),
        ("down_b1", 1, 4096, 14336), ("lm_head_b1", 1, 128256, 4096),
    ]
    for label, M, N, K in shapes:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        wt = w.t()
        for name, fn in [("blaslt", lambda: x @ wt),
                         ("skinny", lambda: ext.skinny_linear(x, w))]:
            for _ in range(5):
                fn()
            torch.cuda.synchronize()
            n = 30
            t0 = _t.perf_counter()
            for _ in range(n):
                fn()
            torch.cuda.synchronize()
            us = (_t.perf_counter() - t0) / n * 1e6
            wbytes = 2 * N * K
            print(f"{label:12s} [{name}] M={M:3d} N={N:6d} K={K:5d}: "
                  f"{us:7.1f} us  W {wbytes/1e6:6.1f} MB  "
                  f"{wbytes/us/1e3:5.2f} TB/s")


def moe_bench():
    """Per-op timing of the dense MoE layer path (Mixtral shapes)."""
    import time as _t

    import torch

    from wva_amd import ops

    E, I, H, B = 8, 14336, 4096, 64
    dt = torch.bfloat16
    dev = "cuda"
    h2 = torch.randn(B, H, device=dev, dtype=dt)
    w_flat = torch.randn(E * 2 * I, H, device=dev, dtype=dt)
    w_down_t = torch.randn(E, I, H, device=dev, dtype=dt)
    act_e = torch.randn(E, B, I, device=dev, dtype=dt)
    w_full = torch.rand(B, E, device=dev, dtype=dt)
    y = torch.randn(E, B, H, device=dev, dtype=dt)
    gate_up = torch.randn(B, E * 2 * I, device=dev, dtype=dt)

    def bench(name, fn, n=10):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = _t.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        print(f"{name:28s} {(_t.perf_counter()-t0)/n*1e3:9.3f} ms")

    bench("gate_up flat linear", lambda: h2 @ w_flat.t())
    bench("silu_mul_fused", lambda: ops.silu_mul_fused(
        gate_up.reshape(B * E, 2 * I)))
    bench("down bmm", lambda: torch.bmm(act_e, w_down_t))
    bench("combine einsum", lambda: torch.einsum("ebh,be->bh", y, w_full))
    bench("combine mul+sum", lambda: (y * w_full.t().unsqueeze(-1)).sum(dim=0))
    bench("act transpose+contig", lambda: act_e.transpose(0, 1).reshape(
        B, E, I).transpose(0, 1).contiguous())
    # the full layer path
    from wva_amd.calibration.moe_model import MIXTRAL_8X7B, _MoELayer
    gen = torch.Generator(device=dev); gen.manual_seed(0)
    layer = _MoELayer(MIXTRAL_8X7B, torch.device(dev), dt, gen)
    from wva_amd.calibration.moe_model import MixtralDecodeModel
    m = MixtralDecodeModel.__new__(MixtralDecodeModel)
    m.cfg = MIXTRAL_8X7B
    logits = h2.float() @ layer.w_router.t().float()
    w, sel = torch.topk(logits, 2, dim=-1)
    w = torch.softmax(w, dim=-1).to(dt)
    bench("full _moe_mlp_dense", lambda: m._moe_mlp_dense(layer, h2, w, sel))
    bench("full _moe_mlp_sparse", lambda: m._moe_mlp_sparse(layer, h2, w, sel), n=3)
