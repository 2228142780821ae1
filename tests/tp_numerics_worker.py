"""Worker: TP=2 decode must match an unsharded TP=1 reference.

Launched by torch.distributed.run from test_tp_cpu.py (gloo/cpu) and
test_tp_gpu.py (nccl=RCCL/cuda). Env: WVA_TP_DEVICE=cpu|cuda,
WVA_TP_BACKEND=gloo|nccl.

The reference model's weights are reassembled from the exact generator
streams the TP shards draw (seed*1000+rank+1 per shard), so the two
engines compute the same function: column-parallel qkv/gate_up shards
concatenate on the row axis (global head / channel order), row-parallel
o/down shards concatenate on the column axis.
"""
import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from wva_amd.calibration.model import TINY, LlamaDecodeModel  # noqa: E402
from wva_amd.calibration.tp_model import TPLlamaDecodeModel  # noqa: E402


def main() -> None:
    device = os.environ.get("WVA_TP_DEVICE", "cpu")
    backend = os.environ.get("WVA_TP_BACKEND", "gloo")
    dist.init_process_group(backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    if device == "cuda":
        torch.cuda.set_device(int(os.environ["LOCAL_RANK"]))

    SEED, B = 7, 4
    tp_model = TPLlamaDecodeModel(
        TINY, max_batch=B, max_seq=64, device=device, seed=SEED
    )
    torch.manual_seed(123)
    tokens = torch.randint(0, TINY.vocab_size, (B,), device=device)
    tp_model.context_lens.zero_()
    logits_tp = tp_model.decode_step(tokens.clone())

    if rank == 0:
        ref = LlamaDecodeModel(TINY, max_batch=B, max_seq=64,
                               device=device, seed=SEED)
        qs = TINY.num_q_heads // world * TINY.head_dim
        ks = TINY.num_kv_heads // world * TINY.head_dim
        inter = TINY.intermediate_size // world
        h = TINY.hidden_size
        for li, layer in enumerate(ref.layers):
            wqkv_parts, wo_parts = [], []
            wgu_g, wgu_u, wdown_parts = [], [], []
            for r in range(world):
                gen = torch.Generator(device=device)
                gen.manual_seed(SEED * 1000 + r + 1)

                def w(rows, cols, gen=gen):
                    return torch.empty(
                        rows, cols, device=device, dtype=torch.bfloat16
                    ).normal_(0.0, 0.02, generator=gen)

                # advance the stream past the earlier layers' draws
                # (generator order in _TPLayer: wqkv, wo, w_gate_up,
                # w_down per layer)
                for _ in range(li):
                    w(qs + 2 * ks, h); w(h, qs)
                    w(2 * inter, h); w(h, inter)
                wqkv = w(qs + 2 * ks, h)
                wo = w(h, qs)
                wgu = w(2 * inter, h)
                wdown = w(h, inter)
                wqkv_parts.append(wqkv)
                wo_parts.append(wo)
                wgu_g.append(wgu[:inter])
                wgu_u.append(wgu[inter:])
                wdown_parts.append(wdown)
            qrows = torch.cat([p[:qs] for p in wqkv_parts], 0)
            krows = torch.cat([p[qs:qs + ks] for p in wqkv_parts], 0)
            vrows = torch.cat([p[qs + ks:] for p in wqkv_parts], 0)
            layer.wqkv.copy_(torch.cat([qrows, krows, vrows], 0))
            layer.wo.copy_(torch.cat(wo_parts, 1))
            layer.w_gate_up.copy_(
                torch.cat([torch.cat(wgu_g, 0), torch.cat(wgu_u, 0)], 0)
            )
            layer.w_down.copy_(torch.cat(wdown_parts, 1))
        # TP lm_head is a fresh first draw of the seed stream — identical
        # to the embed (tied); mirror that in the reference
        ref.lm_head.copy_(ref.embed)
        ref.context_lens.zero_()
        logits_ref = ref.decode_step(tokens.clone())
        err = (logits_tp.float() - logits_ref.float()).abs().max().item()
        scale = logits_ref.float().abs().max().item()
        rel = err / max(scale, 1e-6)
        assert rel < 5e-2, f"TP2 vs TP1 rel err {rel} (abs {err})"
        print(f"TP2_NUMERICS_OK rel={rel:.4g}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
