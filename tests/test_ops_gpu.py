"""GPU numerics tests: each HIP kernel vs the plain fp32 torch reference.

All marked @pytest.mark.gpu — run on an MI355X box via gpurun / the
driver's round-end pass. The HIP extension must be loaded (fail-loud: no
eager fallback on GPU).
"""
import math

import pytest
import torch

from wva_amd import ops

pytestmark = pytest.mark.gpu

B, H = 16, 4096


def tol(ref):
    # bf16 output tolerance relative to fp32 reference
    return dict(atol=2e-2, rtol=2e-2)


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    # The GPU path must use the in-tree HIP extension — no fallback.
    assert ops.extension_available(), "HIP extension _wva_ops not loaded"
    return torch.device("cuda:0")


class TestRMSNorm:
    def test_plain(self, dev):
        x = torch.randn(B, H, device=dev, dtype=torch.bfloat16)
        w = torch.randn(H, device=dev, dtype=torch.bfloat16)
        out = ops.rmsnorm(x, w, None, 1e-5)
        ref, _ = ops.rmsnorm_ref(x, w, None, 1e-5)
        torch.testing.assert_close(out.float(), ref, **tol(ref))

    def test_fused_residual(self, dev):
        x = torch.randn(B, H, device=dev, dtype=torch.bfloat16)
        res = torch.randn(B, H, device=dev, dtype=torch.bfloat16)
        w = torch.randn(H, device=dev, dtype=torch.bfloat16)
        res_before = res.clone()
        out = ops.rmsnorm(x, w, res, 1e-5)
        ref, folded = ops.rmsnorm_ref(x, w, res_before, 1e-5)
        torch.testing.assert_close(out.float(), ref, **tol(ref))
        # residual updated in place to x + residual
        torch.testing.assert_close(res.float(), folded, **tol(folded))

    def test_odd_rows(self, dev):
        x = torch.randn(3, 2048, device=dev, dtype=torch.bfloat16)
        w = torch.ones(2048, device=dev, dtype=torch.bfloat16)
        out = ops.rmsnorm(x, w)
        ref, _ = ops.rmsnorm_ref(x, w)
        torch.testing.assert_close(out.float(), ref, **tol(ref))


class TestRope:
    def test_matches_reference(self, dev):
        T, Hq, Hk, D = 9, 32, 8, 128
        q = torch.randn(T, Hq, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn(T, Hk, D, device=dev, dtype=torch.bfloat16)
        pos = torch.randint(0, 2000, (T,), device=dev, dtype=torch.int32)
        q_ref, k_ref = ops.rope_ref(q, k, pos, 500000.0)
        ops.rope(q, k, pos, 500000.0)
        torch.testing.assert_close(q.float(), q_ref, atol=5e-2, rtol=5e-2)
        torch.testing.assert_close(k.float(), k_ref, atol=5e-2, rtol=5e-2)

    def test_position_zero_identity(self, dev):
        T, Hq, Hk, D = 4, 8, 2, 128
        q = torch.randn(T, Hq, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn(T, Hk, D, device=dev, dtype=torch.bfloat16)
        q0, k0 = q.clone(), k.clone()
        pos = torch.zeros(T, device=dev, dtype=torch.int32)
        ops.rope(q, k, pos)
        torch.testing.assert_close(q.float(), q0.float(), atol=1e-3, rtol=1e-3)


class TestSiluMul:
    def test_matches_reference(self, dev):
        g = torch.randn(B, 14336, device=dev, dtype=torch.bfloat16)
        u = torch.randn(B, 14336, device=dev, dtype=torch.bfloat16)
        out = ops.silu_mul(g, u)
        ref = ops.silu_mul_ref(g, u)
        torch.testing.assert_close(out.float(), ref, **tol(ref))


class TestDecodeAttention:
    @pytest.mark.parametrize("batch,hq,hk,ctx", [
        (1, 32, 8, 17),
        (4, 32, 8, 511),
        (2, 8, 8, 64),     # MHA (G=1)
        (2, 64, 8, 300),   # 70B shape (G=8)
    ])
    def test_matches_reference(self, dev, batch, hq, hk, ctx):
        D, S = 128, 512
        q = torch.randn(batch, hq, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn(batch, hk, S, D, device=dev, dtype=torch.bfloat16)
        v = torch.randn(batch, hk, S, D, device=dev, dtype=torch.bfloat16)
        lens = torch.full((batch,), ctx, device=dev, dtype=torch.int32)
        scale = 1.0 / math.sqrt(D)
        out = ops.gqa_decode_attn(q, k, v, lens, scale)
        ref = ops.gqa_decode_attn_ref(q, k, v, lens, scale)
        torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)

    def test_varied_context_lens(self, dev):
        D, S, batch = 128, 256, 3
        q = torch.randn(batch, 32, D, device=dev, dtype=torch.bfloat16)
        k = torch.randn(batch, 8, S, D, device=dev, dtype=torch.bfloat16)
        v = torch.randn(batch, 8, S, D, device=dev, dtype=torch.bfloat16)
        lens = torch.tensor([1, 100, 256], device=dev, dtype=torch.int32)
        out = ops.gqa_decode_attn(q, k, v, lens)
        ref = ops.gqa_decode_attn_ref(q, k, v, lens, 1.0 / math.sqrt(D))
        torch.testing.assert_close(out.float(), ref, atol=3e-2, rtol=3e-2)


class TestDecodeModel:
    def test_tiny_decode_step(self, dev):
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        model = LlamaDecodeModel(TINY, max_batch=4, max_seq=64)
        model.reset(4, 16)
        tokens = torch.randint(0, TINY.vocab_size, (4,), device=dev)
        logits = model.decode_step(tokens)
        assert logits.shape == (4, TINY.vocab_size)
        assert torch.isfinite(logits.float()).all()
        assert int(model.context_lens[0]) == 17

    def test_calibration_fit(self, dev):
        from wva_amd.calibration.itl_benchmark import calibrate_service_profile
        from wva_amd.calibration.model import TINY

        profile, result = calibrate_service_profile(
            TINY, batch_sizes=[1, 4, 8], context_len=16, max_seq=64, iters=3
        )
        assert result.alpha_ms > 0
        assert profile.num_gpu_blocks > 0


class TestFusedDecodeOps:
    def test_rope_append_kv_matches_unfused(self, dev):
        B, Hq, Hk, D, S = 5, 32, 8, 128, 64
        qkv = torch.randn(B, (Hq + 2 * Hk) * D, device=dev, dtype=torch.bfloat16)
        k_cache = torch.zeros(B, Hk, S, D, device=dev, dtype=torch.bfloat16)
        v_cache = torch.zeros_like(k_cache)
        pos = torch.tensor([0, 3, 10, 31, 63], device=dev, dtype=torch.int32)
        q = ops.rope_append_kv(qkv, k_cache, v_cache, pos, Hq, Hk, 500000.0)

        # unfused reference on CPU
        qkv_c = qkv.cpu()
        kc = torch.zeros(B, Hk, S, D, dtype=torch.bfloat16)
        vc = torch.zeros_like(kc)
        q_ref = ops.rope_append_kv(qkv_c, kc, vc, pos.cpu(), Hq, Hk, 500000.0)
        torch.testing.assert_close(q.float().cpu(), q_ref.float(),
                                   atol=5e-2, rtol=5e-2)
        torch.testing.assert_close(k_cache.float().cpu(), kc.float(),
                                   atol=5e-2, rtol=5e-2)
        torch.testing.assert_close(v_cache.float().cpu(), vc.float(),
                                   atol=1e-3, rtol=1e-3)

    def test_silu_mul_fused_matches_split(self, dev):
        gu = torch.randn(16, 2 * 14336, device=dev, dtype=torch.bfloat16)
        out = ops.silu_mul_fused(gu)
        ref = ops.silu_mul_ref(gu[:, :14336], gu[:, 14336:])
        torch.testing.assert_close(out.float(), ref, atol=2e-2, rtol=2e-2)


class TestMoEDecodeModel:
    def test_tiny_moe_decode_step(self, dev):
        from wva_amd.calibration.moe_model import TINY_MOE, MixtralDecodeModel

        model = MixtralDecodeModel(TINY_MOE, max_batch=4, max_seq=64)
        model.reset(4, 16)
        tokens = torch.randint(0, TINY_MOE.vocab_size, (4,), device=dev)
        logits = model.decode_step(tokens)
        assert logits.shape == (4, TINY_MOE.vocab_size)
        assert torch.isfinite(logits.float()).all()

    def test_moe_matches_cpu_reference(self, dev):
        """Same seed → GPU (HIP kernels) and CPU (fp32 refs) agree."""
        from wva_amd.calibration.moe_model import TINY_MOE, MixtralDecodeModel

        gpu = MixtralDecodeModel(TINY_MOE, max_batch=2, max_seq=32, seed=7)
        cpu = MixtralDecodeModel(
            TINY_MOE, max_batch=2, max_seq=32, device="cpu", seed=7
        )
        # copy GPU weights to CPU model so the comparison isolates kernels
        cpu.embed = gpu.embed.cpu()
        cpu.lm_head = gpu.lm_head.cpu()
        cpu.final_norm = gpu.final_norm.cpu()
        for lc, lg in zip(cpu.layers, gpu.layers):
            for attr in ("input_norm", "post_attn_norm", "wqkv", "wo",
                         "w_router"):
                setattr(lc, attr, getattr(lg, attr).cpu())
            lc.set_expert_weights(
                lg.w_gate_up_stacked.cpu(), lg.w_down_stacked.cpu()
            )
        gpu.context_lens[:2] = 0
        cpu.context_lens[:2] = 0
        tokens = torch.randint(0, TINY_MOE.vocab_size, (2,))
        lg = gpu.decode_step(tokens.to(dev))
        lc = cpu.decode_step(tokens)
        torch.testing.assert_close(
            lg.float().cpu(), lc.float(), atol=0.5, rtol=0.1
        )


class TestTPModel:
    def test_tp1_degenerate_matches_shapes(self, dev):
        """TP=1 (no process group): runs the sharded engine end to end."""
        from wva_amd.calibration.model import TINY
        from wva_amd.calibration.tp_model import TPLlamaDecodeModel

        model = TPLlamaDecodeModel(TINY, max_batch=2, max_seq=32)
        model.reset(2, 8)
        tokens = torch.randint(0, TINY.vocab_size, (2,), device=dev)
        logits = model.decode_step(tokens)
        assert logits.shape == (2, TINY.vocab_size)
        assert torch.isfinite(logits.float()).all()


class TestGraphedDecoder:
    def test_graph_matches_eager(self, dev):
        """hipGraph replay of decode_step == eager decode_step, stepwise."""
        from wva_amd.calibration.graph import GraphedDecoder
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        model = LlamaDecodeModel(TINY, max_batch=4, max_seq=64, seed=3)
        tokens = torch.randint(0, TINY.vocab_size, (4,), device=dev)

        # eager trace from a fixed state
        model.reset(4, 8)
        eager = [model.decode_step(tokens).clone() for _ in range(3)]

        # graphed trace from the same state (reset_to re-seeds; the
        # KV randomization uses the cache tensors' RNG so re-seed the
        # cache deterministically via the same reset path)
        torch.manual_seed(0)
        model.reset(4, 8)
        eager2 = model.decode_step(tokens).clone()
        torch.manual_seed(0)
        dec = GraphedDecoder(model, batch=4, warmup_steps=2)
        dec.reset_to(4, 8)
        graphed = dec.decode_step(tokens).clone()
        torch.testing.assert_close(graphed.float(), eager2.float(),
                                   atol=1e-2, rtol=1e-2)
        # graph advances context state across replays like eager does
        g2 = dec.decode_step(tokens).clone()
        assert not torch.equal(graphed, g2)  # position advanced
        assert torch.isfinite(g2.float()).all()
        del eager

    def test_graph_overrun_guard(self, dev):
        from wva_amd.calibration.graph import GraphedDecoder
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        model = LlamaDecodeModel(TINY, max_batch=2, max_seq=16)
        model.reset(2, 14)
        with pytest.raises(ValueError):
            GraphedDecoder(model, batch=2, warmup_steps=3)


class TestTunedGemms:
    def test_tuned_table_loads(self, dev):
        from wva_amd.ops import enable_tuned_gemms

        assert enable_tuned_gemms() is True
        import torch as _t
        assert _t.cuda.tunable.is_enabled()
        assert not _t.cuda.tunable.tuning_is_enabled()


class TestAttnV4:
    @pytest.mark.parametrize("batch,hq,hk,ctx", [
        (2, 8, 2, 33), (4, 32, 8, 512), (2, 64, 8, 300), (3, 4, 4, 100),
        (2, 8, 1, 2000),  # split path (small grid)
        (1, 4, 1, 20000),  # long-context high-split regime (cap lifted
                           # past 16: ~39 splits at min-tiles 8)
    ])
    def test_v4_matches_reference(self, dev, batch, hq, hk, ctx):
        from wva_amd.ops import _require_ext, gqa_decode_attn_ref

        ext = _require_ext()
        torch.manual_seed(11)
        S = max(ctx + 8, 64)
        q = torch.randn(batch, hq, 128, device=dev, dtype=torch.bfloat16)
        k = torch.randn(batch, hk, S, 128, device=dev, dtype=torch.bfloat16)
        v = torch.randn(batch, hk, S, 128, device=dev, dtype=torch.bfloat16)
        lens = torch.randint(1, ctx + 1, (batch,), device=dev,
                             dtype=torch.int32)
        lens[0] = ctx
        scale = 128 ** -0.5
        out = ext.gqa_decode_attn_v4(q, k, v, lens, scale)
        ref = gqa_decode_attn_ref(
            q.float().cpu(), k.float().cpu(), v.float().cpu(),
            lens.cpu(), scale,
        )
        torch.testing.assert_close(out.float().cpu(), ref, atol=2e-2,
                                   rtol=2e-2)

    def test_v4_matches_v3(self, dev):
        from wva_amd.ops import _require_ext

        ext = _require_ext()
        torch.manual_seed(5)
        q = torch.randn(8, 32, 128, device=dev, dtype=torch.bfloat16)
        k = torch.randn(8, 8, 640, 128, device=dev, dtype=torch.bfloat16)
        v = torch.randn(8, 8, 640, 128, device=dev, dtype=torch.bfloat16)
        lens = torch.full((8,), 640, device=dev, dtype=torch.int32)
        a = ext.gqa_decode_attn(q, k, v, lens, 128 ** -0.5)
        b = ext.gqa_decode_attn_v4(q, k, v, lens, 128 ** -0.5)
        torch.testing.assert_close(a.float(), b.float(), atol=5e-3, rtol=5e-3)


class TestSkinnyLinear:
    @pytest.mark.parametrize("M,N,K", [
        (1, 6144, 4096), (8, 4096, 4096), (16, 28672, 4096),
        (17, 4096, 14336), (33, 128256, 4096), (64, 4096, 4096),
        (64, 28672, 4096), (5, 100, 512), (64, 14336, 512),
    ])
    def test_matches_matmul(self, dev, M, N, K):
        from wva_amd.ops import _require_ext

        ext = _require_ext()
        torch.manual_seed(M * 1000 + N)
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
        y = ext.skinny_linear(x, w)
        ref = (x.float() @ w.float().t())
        torch.testing.assert_close(y.float(), ref, atol=0.5, rtol=2e-2)

    def test_linear_dispatch(self, dev):
        from wva_amd import ops

        x = torch.randn(4, 512, device=dev, dtype=torch.bfloat16)
        w = torch.randn(256, 512, device=dev, dtype=torch.bfloat16)
        y = ops.linear(x, w)
        torch.testing.assert_close(
            y.float(), (x.float() @ w.float().t()), atol=0.5, rtol=2e-2
        )
        # big M falls back to hipBLASLt
        xb = torch.randn(128, 512, device=dev, dtype=torch.bfloat16)
        yb = ops.linear(xb, w)
        assert yb.shape == (128, 256)


class TestAttnV5:
    @pytest.mark.parametrize("batch,hq,hk,ctx", [
        (2, 8, 2, 33), (4, 32, 8, 512), (2, 64, 8, 300), (3, 4, 4, 100),
        (2, 8, 1, 2000),  # split path
    ])
    def test_v5_matches_reference(self, dev, batch, hq, hk, ctx):
        from wva_amd.ops import _require_ext, gqa_decode_attn_ref

        ext = _require_ext()
        torch.manual_seed(17)
        S = max(ctx + 8, 64)
        q = torch.randn(batch, hq, 128, device=dev, dtype=torch.bfloat16)
        k = torch.randn(batch, hk, S, 128, device=dev, dtype=torch.bfloat16)
        v = torch.randn(batch, hk, S, 128, device=dev, dtype=torch.bfloat16)
        lens = torch.randint(1, ctx + 1, (batch,), device=dev,
                             dtype=torch.int32)
        lens[0] = ctx
        scale = 128 ** -0.5
        out = ext.gqa_decode_attn_v5(q, k, v, lens, scale)
        ref = gqa_decode_attn_ref(
            q.float().cpu(), k.float().cpu(), v.float().cpu(),
            lens.cpu(), scale,
        )
        # P is cast to bf16 for the MFMA PV (FA2-standard) — slightly
        # looser tolerance than v4's fp32 PV accumulation
        torch.testing.assert_close(out.float().cpu(), ref, atol=4e-2,
                                   rtol=4e-2)


class TestPrefillGpu:
    def test_prefill_matches_stepwise_on_gpu(self, dev):
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        m1 = LlamaDecodeModel(TINY, max_batch=2, max_seq=32, seed=21)
        m2 = LlamaDecodeModel(TINY, max_batch=2, max_seq=32, seed=21)
        S = 9
        tokens = torch.randint(0, TINY.vocab_size, (2, S), device=dev)
        logits_pre = m1.prefill(tokens)
        m2.context_lens.zero_()
        for s in range(S):
            logits_step = m2.decode_step(tokens[:, s])
        torch.testing.assert_close(
            logits_pre.float(), logits_step.float(), atol=8e-2, rtol=8e-2
        )

    def test_prefill_tps_measured(self, dev):
        from wva_amd.calibration.itl_benchmark import measure_prefill_tps
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        m = LlamaDecodeModel(TINY, max_batch=2, max_seq=64)
        tps = measure_prefill_tps(m, batch=2, seq=64, iters=2)
        assert tps > 0


class TestPrefillAttnKernel:
    @pytest.mark.parametrize("B,hq,hk,S", [
        (2, 8, 2, 64), (1, 32, 8, 512), (2, 4, 4, 100), (1, 64, 8, 333),
        (3, 8, 1, 17),
    ])
    def test_matches_naive_causal(self, dev, B, hq, hk, S):
        from wva_amd import ops

        torch.manual_seed(B * 100 + S)
        S_max = max(S + 8, 64)
        T = B * S
        q = torch.randn(T, hq, 128, device=dev, dtype=torch.bfloat16)
        k = torch.randn(B, hk, S_max, 128, device=dev, dtype=torch.bfloat16)
        v = torch.randn(B, hk, S_max, 128, device=dev, dtype=torch.bfloat16)
        scale = 128 ** -0.5
        out = ops.prefill_attn(q, k, v, B, S, scale)

        # naive fp32 causal reference
        G = hq // hk
        qf = q.float().reshape(B, S, hq, 128)
        ref = torch.empty_like(qf)
        causal = torch.full((S, S), float("-inf"), device=dev).triu(1)
        for h in range(hq):
            kh = k[:, h // G, :S].float()   # [B, S, 128]
            vh = v[:, h // G, :S].float()
            sc = torch.bmm(qf[:, :, h], kh.transpose(1, 2)) * scale + causal
            p = torch.softmax(sc, dim=-1)
            ref[:, :, h] = torch.bmm(p, vh)
        torch.testing.assert_close(
            out.float().reshape(B, S, hq, 128), ref, atol=4e-2, rtol=4e-2
        )


class TestHBMCapacity:
    def test_two_million_token_kv_cache_resident(self, dev):
        """The capacity model's 288 GB claim, physically: allocate the
        Llama-3-8B weights + a ~1.9M-token KV cache (≈248 GB total) and
        run a decode step against the far end of it."""
        import torch

        from wva_amd.calibration.model import LLAMA_3_8B, LlamaDecodeModel

        free, total = torch.cuda.mem_get_info()
        if total < 280 * 2**30:
            pytest.skip("not a 288 GB-class device")
        # 8B: kv_bytes_per_token = 128 KiB -> max_batch*max_seq tokens
        # 64 * 29696 = 1,900,544 tokens ~ 232 GiB of cache
        model = LlamaDecodeModel(LLAMA_3_8B, max_batch=64, max_seq=29696)
        toks = model.max_batch * model.max_seq
        assert toks >= 1_900_000
        model.context_lens[:64] = model.max_seq - 2  # write near the end
        t = torch.randint(0, LLAMA_3_8B.vocab_size, (64,), device=dev)
        logits = model.decode_step(t)
        torch.cuda.synchronize()
        assert torch.isfinite(logits.float()).all()
        used = total - torch.cuda.mem_get_info()[0]
        assert used > 200 * 2**30  # the capacity is genuinely resident


class TestFp8KvGpu:
    def test_fp8_prefill_attn_matches_reference(self, dev):
        from wva_amd import ops

        torch.manual_seed(8)
        B, hq, hk, S = 2, 8, 2, 300
        S_max = S + 8
        T = B * S
        q = torch.randn(T, hq, 128, device=dev, dtype=torch.bfloat16)
        k8 = (torch.randn(B, hk, S_max, 128, device=dev) * 0.7).to(
            torch.float8_e4m3fn)
        v8 = (torch.randn(B, hk, S_max, 128, device=dev) * 0.7).to(
            torch.float8_e4m3fn)
        scale = 128 ** -0.5
        out = ops.prefill_attn(q, k8, v8, B, S, scale)
        G = hq // hk
        qf = q.float().reshape(B, S, hq, 128)
        causal = torch.full((S, S), float("-inf"), device=dev).triu(1)
        for h in range(hq):
            kh = k8[:, h // G, :S].float()
            vh = v8[:, h // G, :S].float()
            sc = torch.bmm(qf[:, :, h], kh.transpose(1, 2)) * scale + causal
            p = torch.softmax(sc, dim=-1)
            ref_h = torch.bmm(p, vh)
            torch.testing.assert_close(
                out.float().reshape(B, S, hq, 128)[:, :, h], ref_h,
                atol=4e-2, rtol=4e-2,
            )

    def test_fp8_attn_matches_reference(self, dev):
        """v4/v5 fp8-KV kernels vs the fp32 reference computed from the
        UPCAST cache contents (isolates kernel error from quantization
        error)."""
        from wva_amd.ops import _require_ext, gqa_decode_attn_ref

        ext = _require_ext()
        torch.manual_seed(3)
        for B, hq, hk, ctx in [(4, 32, 8, 512), (2, 64, 8, 300),
                               (2, 8, 1, 2000)]:
            S = max(ctx + 8, 64)
            q = torch.randn(B, hq, 128, device=dev, dtype=torch.bfloat16)
            k8 = (torch.randn(B, hk, S, 128, device=dev) * 0.7).to(
                torch.float8_e4m3fn)
            v8 = (torch.randn(B, hk, S, 128, device=dev) * 0.7).to(
                torch.float8_e4m3fn)
            lens = torch.full((B,), ctx, device=dev, dtype=torch.int32)
            scale = 128 ** -0.5
            out = ext.gqa_decode_attn(q, k8, v8, lens, scale)
            ref = gqa_decode_attn_ref(
                q.float().cpu(), k8.float().cpu(), v8.float().cpu(),
                lens.cpu(), scale,
            )
            torch.testing.assert_close(
                out.float().cpu(), ref, atol=3e-2, rtol=3e-2
            )

    def test_fp8_rope_append(self, dev):
        from wva_amd import ops

        torch.manual_seed(5)
        B, Hq, Hk, D, S = 3, 8, 2, 128, 32
        qkv = torch.randn(B, (Hq + 2 * Hk) * D, device=dev,
                          dtype=torch.bfloat16)
        k8 = torch.zeros(B, Hk, S, D, device=dev).to(torch.float8_e4m3fn)
        v8 = torch.zeros_like(k8)
        pos = torch.full((B,), 4, device=dev, dtype=torch.int32)
        q = ops.rope_append_kv(qkv, k8, v8, pos, Hq, Hk, theta=500000.0)

        # bf16 reference caches
        kb = torch.zeros(B, Hk, S, D, device=dev, dtype=torch.bfloat16)
        vb = torch.zeros_like(kb)
        q2 = ops.rope_append_kv(qkv, kb, vb, pos, Hq, Hk, theta=500000.0)
        torch.testing.assert_close(q.float(), q2.float())
        torch.testing.assert_close(
            k8[:, :, 4].float(), kb[:, :, 4].float(), atol=8e-2, rtol=8e-2
        )
        torch.testing.assert_close(
            v8[:, :, 4].float(), vb[:, :, 4].float(), atol=8e-2, rtol=8e-2
        )

    def test_fp8_decode_model_end_to_end(self, dev):
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        m = LlamaDecodeModel(TINY, max_batch=2, max_seq=32, seed=9,
                             kv_dtype="fp8")
        m.reset(2, 8)
        t = torch.randint(0, TINY.vocab_size, (2,), device=dev)
        logits = m.decode_step(t)
        assert torch.isfinite(logits.float()).all()
        # fp8 prefill works too (templated flash kernel)
        pre = m.prefill(torch.randint(0, TINY.vocab_size, (1, 4), device=dev))
        assert torch.isfinite(pre.float()).all()

    def test_fp8_weights_decode_gpu(self, dev):
        from wva_amd.calibration.model import TINY, LlamaDecodeModel

        m = LlamaDecodeModel(TINY, max_batch=2, max_seq=32, seed=3,
                             weights_dtype="fp8")
        m.reset(2, 8)
        t = torch.randint(0, TINY.vocab_size, (2,), device=dev)
        logits = m.decode_step(t)
        assert torch.isfinite(logits.float()).all()

    def test_skinny_linear_fp8_matches_dequant(self, dev):
        from wva_amd.ops import _require_ext

        ext = _require_ext()
        torch.manual_seed(2)
        for M, N, K in [(1, 6144, 4096), (8, 4096, 4096), (16, 4096, 14336),
                        (33, 28672, 4096)]:
            x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
            w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
            scale = (w.abs().amax(dim=1).float() / 448.0).clamp_min(1e-12)
            w8 = ((w.float() / scale[:, None]).clamp(-448, 448)
                  .to(torch.float8_e4m3fn).contiguous())
            y = ext.skinny_linear_fp8(x, w8, scale.contiguous())
            ref = x.float() @ (w8.float() * scale[:, None]).t()
            torch.testing.assert_close(y.float(), ref, atol=0.5, rtol=2e-2)


@pytest.mark.gpu
class TestInfernoSizingOnMeasuredParms:
    def test_sized_rate_tracks_measured_saturation(self):
        """Bind the SLO sizer to hardware: calibrate the tiny engine on
        THIS GPU, feed the RAW ITL fit through from_itl_fit (the β
        convention converter), and check the queueing model's rate_max
        against the directly measured saturated throughput — the two
        must agree within modeling tolerance. Guards the ~3× β
        convention bug class with a measurement, not a fixture."""
        from wva_amd.calibration.itl_benchmark import calibrate_service_profile
        from wva_amd.calibration.model import TINY
        from wva_amd.inferno.queue_analyzer import (
            Configuration, QueueAnalyzer, RequestSize, ServiceParms,
        )
        from wva_amd.inferno.types import ServiceParmsSpec

        AVG_IN, AVG_OUT = 100.0, 50.0
        profile, cal = calibrate_service_profile(
            TINY, batch_sizes=[1, 8, 32, 64], context_len=256,
            max_seq=512, iters=5,
        )
        parms = ServiceParmsSpec.from_itl_fit(
            cal.alpha_ms, cal.beta_ms, AVG_IN, AVG_OUT
        )
        qa = QueueAnalyzer(
            Configuration(
                max_batch_size=64, max_queue_size=640,
                service_parms=ServiceParms(
                    alpha=parms.alpha, beta=parms.beta, gamma=parms.gamma
                ),
            ),
            RequestSize(avg_input_tokens=AVG_IN, avg_output_tokens=AVG_OUT),
        )
        # directly measured saturated request rate at batch 64
        measured_rate = 64.0 / (cal.itl_ms[-1] / 1000.0) / AVG_OUT
        # the model's rate_max adds a prefill term the direct number
        # ignores — agree within 35%
        assert qa.rate_max == pytest.approx(measured_rate, rel=0.35), (
            qa.rate_max, measured_rate
        )
        # (the raw-β inflation only bites when β·B is comparable to α —
        # true for 8B at batch 256 but not for TINY at 64; the bug-class
        # regression check with realistic parameters lives in
        # tests/test_inferno_deployable.py on CPU)
