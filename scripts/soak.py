#!/usr/bin/env python3
"""Mixed-workload GPU soak: bf16 Llama-8B (hipGraph), fp8-KV Llama-8B
and Mixtral-8x7B engines RESIDENT TOGETHER in 288 GB HBM3E, decoding in
round-robin bursts for --minutes, while the autoscaler control plane
ticks against the emulated cluster in the same process (the co-resident
production shape: control plane + calibration engines on one host).

Prints one JSON summary (tokens decoded per engine, HBM high-water,
control-plane ticks) and asserts finite logits throughout — the
round-end stability evidence (r1 precedent: 10-min soak, 3.1M tokens).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--minutes", type=float, default=10.0)
    p.add_argument("--batch", type=int, default=64)
    args = p.parse_args()

    import torch

    from wva_amd.calibration.graph import GraphedDecoder
    from wva_amd.calibration.model import LLAMA_3_8B, LlamaDecodeModel
    from wva_amd.calibration.moe_model import MIXTRAL_8X7B, MixtralDecodeModel
    from wva_amd.ops import enable_tuned_gemms

    enable_tuned_gemms()
    B = args.batch
    CTX, MAXSEQ = 512, 1024

    # engine registry: name -> (model, step_fn, batch, reset_ctx, max_seq)
    engines = {}
    bf16 = LlamaDecodeModel(LLAMA_3_8B, max_batch=B, max_seq=MAXSEQ)
    bf16.reset(B, CTX)
    dec = GraphedDecoder(bf16, B, warmup_steps=2)
    dec.reset_to(B, CTX)
    engines["llama8b_bf16_graph"] = (bf16, dec.decode_step, B, CTX, MAXSEQ)

    fp8 = LlamaDecodeModel(LLAMA_3_8B, max_batch=B, max_seq=MAXSEQ,
                           kv_dtype="fp8")
    fp8.reset(B, CTX)
    engines["llama8b_fp8kv"] = (fp8, fp8.decode_step, B, CTX, MAXSEQ)

    moe = MixtralDecodeModel(MIXTRAL_8X7B, max_batch=B, max_seq=MAXSEQ)
    moe.reset(B, CTX)
    engines["mixtral8x7b"] = (moe, moe.decode_step, B, CTX, MAXSEQ)

    # long-context engine: B=4 at 64k context (the lifted split-cap
    # regime) — sustained deep-KV sweeps alongside the short-ctx engines
    LCTX = 65536
    lc = LlamaDecodeModel(LLAMA_3_8B, max_batch=4, max_seq=LCTX + 2048)
    lc.reset(4, LCTX)
    engines["llama8b_longctx64k"] = (lc, lc.decode_step, 4, LCTX, LCTX + 2048)

    # 128k single-sequence engine: the deepest split-KV + 8-wave-merge
    # path under sustained load
    XCTX = 131072
    xl = LlamaDecodeModel(LLAMA_3_8B, max_batch=1, max_seq=XCTX + 2048)
    xl.reset(1, XCTX)
    engines["llama8b_ctx128k"] = (xl, xl.decode_step, 1, XCTX, XCTX + 2048)

    # CPU-side control plane on the emulated cluster, ticking between bursts
    sys.path.insert(0, os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"
    ))
    from test_e2e_emulated import MODEL, NS, make_stack, run_sim

    from wva_amd.emulator.vllm_sim import ServiceProfile

    cluster, sim, app = make_stack(
        replicas=2,
        profile=ServiceProfile(alpha_ms=4.7, beta_ms=0.026,
                               num_gpu_blocks=125_044),
        analyzer="saturation",
    )
    cp_model = sim.model(MODEL, NS)

    tokens = {
        name: torch.randint(0, eng.cfg.vocab_size, (eb,), device="cuda")
        for name, (eng, _, eb, _, _) in engines.items()
    }
    counts = {name: 0 for name in engines}
    ticks = 0
    hbm_peak = 0
    deadline = time.monotonic() + args.minutes * 60.0
    t0 = time.monotonic()
    last_report = t0

    while time.monotonic() < deadline:
        for name, (eng, step, eb, ectx, emax) in engines.items():
            for _ in range(16):
                logits = step(tokens[name])
                counts[name] += eb
            torch.cuda.synchronize()
            assert torch.isfinite(logits.float()).all(), name
            if eng.context_lens[0].item() >= emax - 20:
                eng.reset(eb, ectx)
                if name == "llama8b_bf16_graph":
                    dec.reset_to(eb, ectx)
        # control-plane tick on the CPU between GPU bursts
        run_sim(sim, cp_model, qps=30, seconds=2)
        app.saturation_engine.optimize()
        ticks += 1
        free, total = torch.cuda.mem_get_info()
        hbm_peak = max(hbm_peak, total - free)
        now = time.monotonic()
        if now - last_report > 60:
            last_report = now
            print(f"[soak] t={now - t0:.0f}s tokens={sum(counts.values()):,}"
                  f" hbm={hbm_peak / 2**30:.1f}GiB", file=sys.stderr)

    elapsed = time.monotonic() - t0
    summary = {
        "soak_minutes": round(elapsed / 60.0, 2),
        "device": torch.cuda.get_device_name(0),
        "tokens": counts,
        "total_tokens": sum(counts.values()),
        "tokens_per_s": round(sum(counts.values()) / elapsed),
        "hbm_high_water_gib": round(hbm_peak / 2**30, 1),
        "control_plane_ticks": ticks,
        "engines_resident": list(engines),
    }
    print(json.dumps(summary))
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/soak_r2.json", "w") as f:
        json.dump(summary, f, indent=2)


if __name__ == "__main__":
    main()
