"""GPU numerics tests: every gfx950 HIP kernel vs the plain-PyTorch fp32
reference of the same op (SURVEY.md §4(d) strategy). Run via gpurun:
  python -m pytest tests/test_ops_gpu.py -m gpu -x -q
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from datatunerx_amd import ops
    from datatunerx_amd.ops import reference as ref
    DEV = torch.device("cuda:0")
    assert ops.have_ext(), "HIP extension must be built for GPU tests"

torch.manual_seed(0)


def mk(*shape, dtype=torch.bfloat16, scale=1.0):
    return (torch.randn(*shape, device=DEV, dtype=torch.float32)
            .mul(scale).to(dtype))


def assert_close(got, want, rtol=2e-2, name=""):
    got, want = got.float(), want.float()
    denom = want.abs().max().clamp(min=1e-3)
    err = (got - want).abs().max() / denom
    assert err < rtol, f"{name}: rel err {err:.4f} (max {denom:.3f})"


# ---------------------------------------------------------------- rmsnorm
@pytest.mark.parametrize("M,H", [(512, 4096), (33, 128), (2048, 5120)])
def test_rmsnorm_fwd_bwd(M, H):
    x, w = mk(M, H), mk(H)
    y, inv = ops.rmsnorm_fwd(x, w, 1e-5)
    y_ref, inv_ref = ref.rmsnorm_fwd(x.cpu(), w.cpu(), 1e-5)
    assert_close(y.cpu(), y_ref, name="rmsnorm y")
    assert_close(inv.cpu(), inv_ref, name="rmsnorm inv")
    dy = mk(M, H)
    dx, dw = ops.rmsnorm_bwd(dy, x, w, inv)
    dx_ref, dw_ref = ref.rmsnorm_bwd(dy.cpu(), x.cpu(), w.cpu(), inv_ref)
    assert_close(dx.cpu(), dx_ref, name="rmsnorm dx")
    assert_close(dw.cpu(), dw_ref, rtol=3e-2, name="rmsnorm dw")


# ------------------------------------------------------------------- rope
@pytest.mark.parametrize("D", [128, 64])
def test_rope(D):
    B, S, H = 2, 130, 4
    cos, sin = ref.rope_tables(256, D, device=DEV)
    x = mk(B, S, H, D)
    y = ops.rope_fwd(x, cos, sin, pos0=3)
    y_ref = ref.rope_fwd(x.cpu(), cos.cpu(), sin.cpu(), 3)
    assert_close(y.cpu(), y_ref, name="rope fwd")
    dy = mk(B, S, H, D)
    dx = ops.rope_bwd(dy, cos, sin, pos0=3)
    dx_ref = ref.rope_bwd(dy.cpu(), cos.cpu(), sin.cpu(), 3)
    assert_close(dx.cpu(), dx_ref, name="rope bwd")


# ----------------------------------------------------------------- swiglu
def test_swiglu():
    g, u = mk(1024, 1408), mk(1024, 1408)
    out = ops.swiglu_fwd(g, u)
    assert_close(out.cpu(), ref.swiglu_fwd(g.cpu(), u.cpu()), name="swiglu")
    d = mk(1024, 1408)
    dg, du = ops.swiglu_bwd(d, g, u)
    dg_r, du_r = ref.swiglu_bwd(d.cpu(), g.cpu(), u.cpu())
    assert_close(dg.cpu(), dg_r, name="swiglu dg")
    assert_close(du.cpu(), du_r, name="swiglu du")


# ------------------------------------------------------------------- xent
@pytest.mark.parametrize("N,V", [(512, 32000), (37, 512)])
def test_xent(N, V):
    logits = mk(N, V, scale=3.0)
    targets = torch.randint(0, V, (N,), device=DEV)
    targets[::7] = -100
    loss, lse = ops.softmax_xent_fwd(logits, targets)
    loss_r, lse_r = ref.softmax_xent_fwd(logits.cpu(), targets.cpu())
    assert_close(loss.cpu(), loss_r, rtol=1e-2, name="xent loss")
    assert_close(lse.cpu(), lse_r, rtol=1e-2, name="xent lse")
    dloss = torch.rand(N, device=DEV)
    dl = ops.softmax_xent_bwd(logits, targets, lse, dloss)
    dl_r = ref.softmax_xent_bwd(logits.cpu(), targets.cpu(), lse_r,
                                dloss.cpu())
    assert_close(dl.cpu(), dl_r, rtol=3e-2, name="xent dlogits")


# ------------------------------------------------------------------- lora
@pytest.mark.parametrize("M,K,r", [(1024, 4096, 8), (513, 4096, 16),
                                   (256, 11008, 8), (512, 2048, 64),
                                   (256, 5120, 8)])   # 13B K (odd split)
def test_lora_contract(M, K, r):
    x, w = mk(M, K, scale=0.3), mk(r, K, scale=0.3)
    t = ops.lora_contract(x, w)
    t_ref = ref.lora_contract(x.cpu(), w.cpu())
    assert_close(t.cpu(), t_ref, name="lora contract")


def test_lora_rng_dropout():
    """seed-mode (in-kernel counter-based RNG) must produce EXACTLY the
    bits of the materialized dropout_mask tensor, in all three kernels,
    and the GPU mask must match the CPU splitmix64 twin bit-for-bit."""
    M, K, N, r = 512, 2048, 2048, 8
    seed, keep = 12345, 0.9
    x, a = mk(M, K, scale=0.3), mk(r, K, scale=0.3)
    mask = ops.dropout_mask(M, K, seed, keep, x)
    # GPU mask == CPU splitmix64 twin (bit-exact)
    mask_cpu = ref.dropout_mask(M, K, seed, keep)
    assert torch.equal(mask.cpu(), mask_cpu), "GPU/CPU RNG mismatch"
    rate = (mask > 0).float().mean().item()
    assert abs(rate - keep) < 0.02

    t_rng = ops.lora_contract(x, a, seed=seed, keep=keep)
    t_mask = ops.lora_contract(x, a, mask)
    assert torch.equal(t_rng, t_mask), "contract rng != mask"

    dt = torch.randn(M, r, device=DEV)
    dw_rng = ops.lora_wgrad(dt, x, 0.5, seed=seed, keep=keep)
    dw_mask = ops.lora_wgrad(dt, x, 0.5, mask)
    assert torch.equal(dw_rng, dw_mask), "wgrad rng != mask"

    y = mk(M, N)
    y2 = y.clone()
    b = mk(N, r, scale=0.3)
    t2 = torch.randn(M, r, device=DEV)
    ops.lora_expand_add(y, t2, b, 0.5, seed=seed, keep=keep)
    mask_y = ops.dropout_mask(M, N, seed, keep, y2)
    ops.lora_expand_add(y2, t2, b, 0.5, mask_y)
    assert torch.equal(y, y2), "expand rng != mask"


def _mk_mask(M, K, keep=0.9):
    m = torch.empty(M, K, device=DEV, dtype=torch.bfloat16)
    m.bernoulli_(keep).mul_(1.0 / keep)
    return m


def test_lora_masked_fused_dropout():
    """Fused-dropout variants (mask inside contract/expand/wgrad) match
    the reference with the same mask."""
    M, K, N, r = 512, 2048, 1024, 8
    x, a = mk(M, K, scale=0.3), mk(r, K, scale=0.3)
    mask = _mk_mask(M, K)
    t = ops.lora_contract(x, a, mask)
    t_ref = ref.lora_contract(x.cpu(), a.cpu(), mask.cpu())
    assert_close(t.cpu(), t_ref, name="masked contract")

    dt = torch.randn(M, r, device=DEV)
    dw = ops.lora_wgrad(dt, x, 0.5, mask)
    dw_ref = ref.lora_wgrad(dt.cpu(), x.cpu(), 0.5, mask.cpu())
    assert_close(dw.cpu(), dw_ref, name="masked wgrad")

    y = mk(M, N)
    y0 = y.clone()
    b = mk(N, r, scale=0.3)
    t2 = torch.randn(M, r, device=DEV)
    mask_y = _mk_mask(M, N)
    ops.lora_expand_add(y, t2, b, 0.5, mask_y)
    y_ref = ref.lora_expand_add(y0.cpu().clone(), t2.cpu(), b.cpu(), 0.5,
                                mask_y.cpu())
    assert_close(y.cpu(), y_ref, name="masked expand")


@pytest.mark.parametrize("M,N,r", [(1024, 4096, 8), (511, 1024, 16)])
def test_lora_expand_add(M, N, r):
    y = mk(M, N)
    y0 = y.clone()
    t = torch.randn(M, r, device=DEV)
    w = mk(N, r, scale=0.3)
    ops.lora_expand_add(y, t, w, 0.5)
    y_ref = ref.lora_expand_add(y0.cpu().clone(), t.cpu(), w.cpu(), 0.5)
    assert_close(y.cpu(), y_ref, name="lora expand")


@pytest.mark.parametrize("M,K,r", [(1024, 4096, 8), (512, 2048, 16)])
def test_lora_wgrad(M, K, r):
    t = torch.randn(M, r, device=DEV)
    x = mk(M, K, scale=0.3)
    dw = ops.lora_wgrad(t, x, 0.7)
    dw_ref = ref.lora_wgrad(t.cpu(), x.cpu(), 0.7)
    assert_close(dw.cpu(), dw_ref, name="lora wgrad")


# ------------------------------------------------------------------ adamw
def test_adamw():
    n = 4096
    master = torch.randn(n, device=DEV)
    p = master.to(torch.bfloat16)
    g = torch.randn(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    mc, pc = master.cpu().clone(), p.cpu().clone()
    gc, mmc, vvc = g.cpu(), m.cpu().clone(), v.cpu().clone()
    for step in (1, 2, 3):
        ops.adamw_step(p, master, g, m, v, 1e-3, 0.9, 0.999, 1e-8, 0.01,
                       step)
        ref.adamw_step(pc, mc, gc, mmc, vvc, 1e-3, 0.9, 0.999, 1e-8, 0.01,
                       step)
    assert_close(master.cpu(), mc, rtol=1e-5, name="adamw master")
    assert_close(m.cpu(), mmc, rtol=1e-5, name="adamw m")


def test_l2_norm():
    x = torch.randn(1 << 20, device=DEV)
    got = ops.l2_norm(x)
    want = ref.l2_norm(x.cpu())
    assert abs(float(got) - float(want)) / float(want) < 1e-4


# -------------------------------------------------------------- attention
@pytest.mark.parametrize("B,Hq,Hkv,S,Skv,D,causal", [
    (2, 4, 4, 128, 128, 128, True),
    (2, 4, 4, 128, 128, 128, False),
    (1, 8, 2, 256, 256, 128, True),       # GQA
    (2, 4, 4, 100, 100, 128, True),       # ragged S
    (1, 2, 2, 64, 192, 128, True),        # Skv > S (cache decode shape)
    (2, 4, 4, 128, 128, 64, True),        # D=64
    (1, 4, 4, 1024, 1024, 128, True),     # training shape
])
def test_attn_fwd(B, Hq, Hkv, S, Skv, D, causal):
    q = mk(B, S, Hq, D, scale=0.5)
    k = mk(B, Skv, Hkv, D, scale=0.5)
    v = mk(B, Skv, Hkv, D, scale=0.5)
    o, lse = ops.attn_fwd(q, k, v, causal, 1.0 / math.sqrt(D))
    o_ref, lse_ref = ref.attn_fwd(q.cpu(), k.cpu(), v.cpu(), causal)
    assert_close(o.cpu(), o_ref, name="attn o")
    assert_close(lse.cpu(), lse_ref, rtol=1e-2, name="attn lse")


@pytest.mark.parametrize("B,Hq,Hkv,S,D,causal", [
    (2, 4, 4, 128, 128, True),
    (2, 4, 4, 128, 128, False),
    (1, 8, 2, 256, 128, True),
    (2, 2, 2, 100, 128, True),
    (1, 4, 4, 512, 64, True),
    (1, 2, 2, 1024, 128, True),           # training shape
])
def test_attn_bwd(B, Hq, Hkv, S, D, causal):
    q = mk(B, S, Hq, D, scale=0.5)
    k = mk(B, S, Hkv, D, scale=0.5)
    v = mk(B, S, Hkv, D, scale=0.5)
    o, lse = ops.attn_fwd(q, k, v, causal, 1.0 / math.sqrt(D))
    do = mk(B, S, Hq, D, scale=0.5)
    dq, dk, dv = ops.attn_bwd(q, k, v, o, do, lse, causal,
                              1.0 / math.sqrt(D))
    o_ref, lse_ref = ref.attn_fwd(q.cpu(), k.cpu(), v.cpu(), causal)
    dq_r, dk_r, dv_r = ref.attn_bwd(q.cpu(), k.cpu(), v.cpu(), o_ref,
                                    do.cpu(), lse_ref, causal)
    assert_close(dq.cpu(), dq_r, rtol=3e-2, name="attn dq")
    assert_close(dk.cpu(), dk_r, rtol=3e-2, name="attn dk")
    assert_close(dv.cpu(), dv_r, rtol=3e-2, name="attn dv")


@pytest.mark.parametrize("B,S,H,D", [(2, 128, 4, 128), (1, 100, 2, 64)])
def test_transpose_sd(B, S, H, D):
    from datatunerx_amd.ops import _dtx_hip
    x = mk(B, S, H, D)
    xt = _dtx_hip.transpose_sd(x)
    want = x.permute(0, 2, 3, 1).contiguous()
    assert torch.equal(xt, want)


# --------------------------------------------- end-to-end GPU train step
def test_tiny_train_step_gpu():
    from datatunerx_amd.data.dataset import SFTDataset
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.train.trainer import SFTTrainer, TrainerConfig
    cfg = LlamaConfig(vocab_size=1024, hidden_size=256,
                      intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=2, num_key_value_heads=2,
                      max_position_embeddings=512)
    with torch.device(DEV):
        model = LlamaForCausalLM(cfg, lora=True, dtype=torch.bfloat16)
    model.init_random()
    ds = SFTDataset.synthetic(32, 256, cfg.vocab_size)
    tr = SFTTrainer(model, ds,
                    TrainerConfig(output_dir="gpurun_out/test_out",
                                  max_steps=8, micro_batch_size=4,
                                  logging_steps=0, learning_rate=1e-3),
                    device=DEV)
    it = iter(tr.train_loader)
    losses = [tr.train_step([next(it)]) for _ in range(8)]
    assert all(l == l for l in losses), f"NaN in {losses}"
    assert sum(losses[-2:]) / 2 < sum(losses[:2]) / 2, losses


def test_serve_engine_decode_gpu():
    """KV-cache decode on GPU: prefill + per-token attention with
    Skv > S through the BSHD kernels; greedy generation is finite and
    deterministic."""
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.serve.engine import InferenceEngine
    cfg = LlamaConfig(vocab_size=512, hidden_size=256,
                      intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=2, num_key_value_heads=2,
                      max_position_embeddings=256)
    with torch.device(DEV):
        model = LlamaForCausalLM(cfg, lora=False, dtype=torch.bfloat16)
    model.init_random(seed=3)
    model.eval()
    eng = InferenceEngine(model, device=DEV)          # hipGraph decode
    out1 = eng.generate(list(range(4, 20)), max_new_tokens=8)
    out2 = eng.generate(list(range(4, 20)), max_new_tokens=8)
    assert out1 == out2
    assert 0 < len(out1) <= 8
    assert eng._graphed is not None, "graph decode did not engage"
    # graph replay must match the eager decode path token-for-token
    eager = InferenceEngine(model, device=DEV, graph_decode=False)
    out3 = eager.generate(list(range(4, 20)), max_new_tokens=8)
    assert out1 == out3
    ppl = eng.perplexity(["hello world", "the quick brown fox"])
    assert 0 < ppl < float("inf")
    # streaming yields the same tokens as batch generation (graph mode)
    streamed = list(eng.generate_stream(list(range(4, 20)),
                                        max_new_tokens=8))
    assert streamed == out1


@pytest.mark.parametrize("B,Hq,Hkv,Skv,D", [
    (1, 32, 32, 1024, 128),
    (2, 8, 2, 777, 128),       # GQA + ragged cache
    (1, 4, 4, 300, 64),
])
def test_attn_decode(B, Hq, Hkv, Skv, D):
    """Flash-decoding S=1 path vs the reference attention."""
    q = mk(B, 1, Hq, D, scale=0.5)
    k = mk(B, Skv, Hkv, D, scale=0.5)
    v = mk(B, Skv, Hkv, D, scale=0.5)
    o, lse = ops.attn_fwd(q, k, v, True, 1.0 / math.sqrt(D))
    o_ref, lse_ref = ref.attn_fwd(q.cpu(), k.cpu(), v.cpu(), True)
    assert_close(o.cpu(), o_ref, name="decode o")
    assert_close(lse.cpu().reshape(-1), lse_ref.reshape(-1), rtol=1e-2,
                 name="decode lse")


@pytest.mark.parametrize("N,K", [(4096, 4096), (32000, 4096),
                                 (11008, 4096), (120, 64)])
def test_gemv(N, K):
    from datatunerx_amd.ops import _dtx_hip
    x = mk(1, K, scale=0.3)
    w = mk(N, K, scale=0.3)
    y = _dtx_hip.gemv(x, w)
    want = torch.nn.functional.linear(x.float(), w.float())
    assert_close(y.cpu(), want.cpu(), name="gemv")


# ------------------------------------------------------- base MFMA GEMM
@pytest.mark.parametrize("M,N,K", [
    (512, 512, 128),          # minimal tile coverage
    (1024, 1024, 4096),       # square projection-ish
    (777, 1024, 4096),        # M tail (SRSRC clamp + store predicate)
    (2048, 11008, 4096),      # gate/up shape family
    (1024, 4096, 11008),      # down-proj (big K)
    (384, 1280, 192),         # odd-but-supported dims
])
def test_gemm_nt(M, N, K):
    a, b = mk(M, K), mk(N, K)
    c = ops.gemm_nt(a, b)
    want = a.float() @ b.float().t()
    assert_close(c, want, name=f"gemm_nt {M}x{N}x{K}")


def test_gemm_nt_src_fused():
    M, N, K = 777, 1024, 512
    a, b, src = mk(M, K), mk(N, K), mk(M, N)
    c = ops.gemm_nt(a, b, src)
    want = a.float() @ b.float().t() + src.float()
    assert_close(c, want, name="gemm_nt+src")


def test_frozen_gemm_autograd():
    from datatunerx_amd.ops.autograd import FrozenGemm
    M, N, K = 512, 1024, 512
    x = mk(M, K)
    x.requires_grad_(True)
    w = mk(N, K)
    wt = w.t().contiguous()
    res = mk(M, N)
    res.requires_grad_(True)
    y = FrozenGemm.apply(x, w, wt, res)
    dy = mk(M, N)
    y.backward(dy)
    want_y = x.detach().float() @ w.float().t() + res.detach().float()
    want_dx = dy.float() @ w.float()
    assert_close(y, want_y, name="frozen_gemm y")
    assert_close(x.grad, want_dx, name="frozen_gemm dx")
    assert_close(res.grad, dy, name="frozen_gemm dres")


# ------------------------------------------------- chunked fused CE
def test_xent_chunk_kernels():
    M, V = 513, 1024
    logits = mk(M, V)
    targets = torch.randint(0, 2 * V, (M,), device=DEV)  # half outside
    targets[::7] = -100
    m = torch.full((M,), -3.4e38, device=DEV)
    l = torch.zeros(M, device=DEV)
    tgt = torch.zeros(M, device=DEV)
    v0 = 512
    ops.xent_lse_merge(logits, targets, m, l, tgt, v0)
    m2 = torch.full((M,), -3.4e38, dtype=torch.float32)
    l2 = torch.zeros(M)
    t2 = torch.zeros(M)
    ref.xent_lse_merge(logits.cpu(), targets.cpu(), m2, l2, t2, v0)
    assert_close(m.cpu(), m2, name="chunk m")
    assert_close(l.cpu(), l2, name="chunk l")
    assert_close(tgt.cpu(), t2, name="chunk tgt")

    lse = m + l.log()
    dl = ops.xent_dlogits(logits, targets, lse, v0)
    dl_ref = ref.xent_dlogits(logits.cpu(), targets.cpu(), lse.cpu(), v0)
    assert_close(dl.cpu(), dl_ref, name="chunk dlogits")


def test_fused_linear_ce_gpu_matches_unfused():
    from datatunerx_amd.ops.autograd import (FusedLinearCrossEntropy,
                                             cross_entropy,
                                             fused_linear_cross_entropy)
    M, V, E = 777, 2048, 256
    h = mk(M, E)
    h.requires_grad_(True)
    w = mk(V, E)
    t = torch.randint(0, V, (M,), device=DEV)
    t[::9] = -100
    old = FusedLinearCrossEntropy.CHUNK
    FusedLinearCrossEntropy.CHUNK = 512
    try:
        loss = fused_linear_cross_entropy(h, w, t)
        loss.backward()
        g1 = h.grad.clone()
        h.grad = None
        loss2 = cross_entropy(torch.nn.functional.linear(h, w), t)
        loss2.backward()
        assert abs(float(loss.detach()) - float(loss2.detach())) < 2e-3 * max(
            1.0, abs(float(loss2)))
        assert_close(g1, h.grad, rtol=3e-2, name="fused-ce dx")
    finally:
        FusedLinearCrossEntropy.CHUNK = old


# ----------------------------------------------------- weight dequant
def test_dequant_kernels_match_reference():
    from datatunerx_amd.models.quant import (dequantize_int4,
                                             dequantize_int8,
                                             quantize_int4, quantize_int8)
    w = mk(512, 1024)
    q8, s8 = quantize_int8(w.cpu())
    got8 = ops.dequant_int8(q8.to(DEV), s8.to(DEV))
    assert_close(got8.cpu(), dequantize_int8(q8, s8, torch.float32),
                 name="int8 dequant")
    q4, s4 = quantize_int4(w.cpu())
    got4 = ops.dequant_int4(q4.to(DEV), s4.to(DEV))
    assert_close(got4.cpu(), dequantize_int4(q4, s4, torch.float32),
                 name="int4 dequant")


@pytest.mark.parametrize("bits", [8, 4])
def test_quantized_train_step_gpu(bits):
    """One quantized LoRA train step on GPU: the base dequant runs the
    HIP kernels and the step produces finite loss + adapter grads
    (VERDICT r1 item 8)."""
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.models.quant import quantize_model_
    torch.manual_seed(0)
    # head_dim 64 (the attention kernels support D in {64, 128})
    cfg = LlamaConfig(vocab_size=512, hidden_size=256,
                      intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=4,
                      max_position_embeddings=256, lora_r=8)
    model = LlamaForCausalLM(cfg, lora=True,
                             dtype=torch.bfloat16).init_random()
    model = model.to(DEV)
    n_q = quantize_model_(model, bits=bits)
    assert n_q > 0
    ids = torch.randint(3, cfg.vocab_size, (2, 64), device=DEV)
    labels = ids.clone()
    loss = model(ids, labels=labels)
    loss.backward()
    assert torch.isfinite(loss)
    for n, p in model.trainable_parameters():
        if "lora_A" in n:
            assert p.grad is not None and torch.isfinite(p.grad).all(), n


# --------------------------------------------- batched ragged decode
def test_batched_decode_matches_sequential_gpu():
    """GPU batched ragged decode (per-row rope positions + per-row
    cache lengths through the HIP kernels) == per-request generation."""
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.serve.engine import InferenceEngine
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=512, hidden_size=256,
                      intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2,
                      max_position_embeddings=256)
    with torch.device(DEV):
        model = LlamaForCausalLM(cfg, lora=False, dtype=torch.bfloat16)
    model.init_random()
    model.eval()
    eng = InferenceEngine(model, template="vanilla", device=DEV,
                          graph_decode=False)
    prompts = [eng.tok.encode("hello world", add_special_tokens=True),
               eng.tok.encode("a much longer prompt with many words in",
                              add_special_tokens=True),
               eng.tok.encode("x", add_special_tokens=True),
               eng.tok.encode("short one", add_special_tokens=True)]
    batched = eng.generate_batch(prompts, max_new_tokens=16)
    seq = [eng.generate(p, max_new_tokens=16) for p in prompts]
    assert batched == seq


@pytest.mark.gpu
def test_qwen_style_train_step_gpu():
    """Qwen2/3-style geometry on hardware: qkv bias adds + per-head
    qk-norm (rmsnorm HIP kernel on [B*S*H, D] rows) through a full
    LoRA train step — loss finite and decreasing."""
    from datatunerx_amd.data.dataset import SFTDataset
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.train.trainer import SFTTrainer, TrainerConfig
    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=1024, hidden_size=256,
                      intermediate_size=512, num_hidden_layers=2,
                      num_attention_heads=4, num_key_value_heads=2,
                      head_dim_override=64, max_position_embeddings=512,
                      attention_bias=True, qk_norm=True)
    with torch.device(DEV):
        model = LlamaForCausalLM(cfg, lora=True, dtype=torch.bfloat16)
    model.init_random()
    with torch.no_grad():
        for lyr in model.layers:
            for nm in ("q_bias", "k_bias", "v_bias"):
                getattr(lyr.self_attn, nm).normal_(0, 0.05)
            lyr.self_attn.q_norm.uniform_(0.5, 1.5)
            lyr.self_attn.k_norm.uniform_(0.5, 1.5)
    ds = SFTDataset.synthetic(32, 256, cfg.vocab_size)
    tr = SFTTrainer(model, ds,
                    TrainerConfig(output_dir="gpurun_out/test_out",
                                  max_steps=8, micro_batch_size=4,
                                  logging_steps=0, learning_rate=1e-3),
                    device=DEV)
    it = iter(tr.train_loader)
    losses = [tr.train_step([next(it)]) for _ in range(8)]
    assert all(l == l for l in losses), f"NaN in {losses}"
    assert sum(losses[-2:]) / 2 < sum(losses[:2]) / 2, losses


@pytest.mark.gpu
def test_qk_norm_matches_reference_gpu():
    """The qk-norm rmsnorm call on [B,S,H,D] head rows matches the fp32
    torch reference on GPU (the per-head-row use of the kernel)."""
    from datatunerx_amd.ops import reference as ref
    from datatunerx_amd.ops.autograd import rmsnorm
    torch.manual_seed(1)
    x = torch.randn(2, 33, 4, 64, device=DEV).to(torch.bfloat16)
    x.requires_grad_(True)
    w = (torch.rand(64, device=DEV) + 0.5).to(torch.bfloat16)
    y = rmsnorm(x, w, 1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    xf = x.detach().float().cpu().requires_grad_(True)
    yr, _ = ref.rmsnorm_fwd(xf, w.float().cpu(), 1e-6)
    yr.backward(g.float().cpu())
    assert torch.allclose(y.float().cpu(), yr, atol=3e-2, rtol=3e-2)
    assert torch.allclose(x.grad.float().cpu(), xf.grad, atol=3e-2,
                          rtol=3e-2)


@pytest.mark.gpu
def test_per_row_cross_entropy_gpu():
    """Unreduced CE (the DPO building block): per-row losses and the
    per-row-weighted backward match the fp32 torch reference."""
    import torch.nn.functional as F

    from datatunerx_amd.ops.autograd import per_row_cross_entropy
    torch.manual_seed(0)
    N, V = 64, 512
    x = torch.randn(N, V, device=DEV).to(torch.bfloat16).requires_grad_(True)
    t = torch.randint(0, V, (N,), device=DEV)
    t[::5] = -100
    loss = per_row_cross_entropy(x, t)
    w = torch.randn(N, device=DEV)               # per-row upstream grads
    (loss * w).sum().backward()

    xf = x.detach().float().cpu().requires_grad_(True)
    tc = t.cpu()
    m = tc != -100
    ref = F.cross_entropy(xf, tc.clamp(min=0), reduction="none") * m
    (ref * w.cpu()).sum().backward()
    assert torch.allclose(loss.cpu(), ref.detach(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(x.grad.float().cpu(), xf.grad, atol=3e-2,
                          rtol=3e-2)


@pytest.mark.gpu
def test_dpo_train_step_gpu(tmp_path):
    """One DPO step on hardware: adapters-off reference pass + per-row
    xent backward; loss starts at -logsigmoid(0) and adapters move."""
    from datatunerx_amd.data.preference import PreferenceDataset
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.train.trainer import DPOTrainer, TrainerConfig
    torch.manual_seed(0)
    cfg = LlamaConfig.mini(lora_dropout=0.0)
    with torch.device(DEV):
        model = LlamaForCausalLM(cfg, lora=True, dtype=torch.bfloat16)
    model.init_random(seed=2)
    ds = PreferenceDataset.synthetic(16, 64, cfg.vocab_size, seed=4)
    tr = DPOTrainer(model, ds, TrainerConfig(
        output_dir=str(tmp_path), max_steps=4, micro_batch_size=4,
        logging_steps=0, learning_rate=5e-3), device=DEV, beta=0.5)
    it = iter(tr.train_loader)
    losses = [tr.train_step([next(it)]) for _ in range(4)]
    assert abs(losses[0] - 0.693) < 0.02, losses   # ref == policy at t0
    assert all(l == l for l in losses), losses     # finite
    # adapters moved away from zero (base stays the reference)
    bsum = sum(float(p.abs().sum()) for n, p in model.named_parameters()
               if "lora_B" in n)
    assert bsum > 0, "adapters did not update"
    # after updates the policy diverges from the reference: the DPO
    # margin is nonzero (short bf16 horizon: direction only)
    assert tr.last_margin != 0.0
