"""Real-checkpoint contract: load an HF-format Llama
checkpoint + tokenizer from a LOCAL directory and match HF transformers'
logits. transformers is a TEST-ONLY dependency (creates the fixture and
the reference logits); the loader itself uses safetensors/tokenizers/
sentencepiece only."""

import json
import os

import pytest
import torch

from datatunerx_amd.models import LlamaForCausalLM
from datatunerx_amd.models.hf_io import (HFTokenizer, is_hf_model_dir,
                                         load_hf_config, load_hf_weights,
                                         load_tokenizer)

transformers = pytest.importorskip("transformers")


@pytest.fixture(scope="module")
def hf_dir(tmp_path_factory):
    d = str(tmp_path_factory.mktemp("hfmodel"))
    cfg = transformers.LlamaConfig(
        vocab_size=320, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, rms_norm_eps=1e-5, rope_theta=10000.0,
        attn_implementation="eager")
    torch.manual_seed(3)
    m = transformers.LlamaForCausalLM(cfg)
    m.save_pretrained(d, safe_serialization=True)

    # build a real BPE tokenizer.json locally (no network)
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers
    tok = Tokenizer(models.BPE(unk_token="<unk>"))
    tok.pre_tokenizer = pre_tokenizers.Whitespace()
    trainer = trainers.BpeTrainer(
        vocab_size=320, special_tokens=["<unk>", "<s>", "</s>"])
    tok.train_from_iterator(
        ["the quick brown fox jumps over the lazy dog",
         "llama models fine tune with low rank adapters",
         "hello world example text for byte pair merges"] * 30, trainer)
    tok.save(os.path.join(d, "tokenizer.json"))
    return d


def test_hf_config_parse(hf_dir):
    assert is_hf_model_dir(hf_dir)
    cfg = load_hf_config(hf_dir)
    assert cfg.vocab_size == 320 and cfg.hidden_size == 64
    assert cfg.num_key_value_heads == 2 and cfg.num_hidden_layers == 2


def test_hf_weights_match_transformers_logits(hf_dir):
    cfg = load_hf_config(hf_dir)
    ours = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    n = load_hf_weights(ours, hf_dir)
    assert n >= 2 * 9 + 3            # per-layer tensors + embed/norm/head
    ours.eval()

    ref = transformers.LlamaForCausalLM.from_pretrained(
        hf_dir, torch_dtype=torch.float32, attn_implementation="eager")
    ref.eval()

    torch.manual_seed(0)
    ids = torch.randint(0, 320, (2, 17))
    with torch.no_grad():
        got = ours(ids)
        want = ref(ids).logits
    assert torch.allclose(got, want, atol=2e-4, rtol=1e-3), \
        (got - want).abs().max()


def test_hf_weights_sharded_index(hf_dir, tmp_path):
    """Sharded checkpoints (model-0000x-of-0000y.safetensors + index)."""
    import shutil

    from safetensors.torch import load_file, save_file
    d = str(tmp_path / "sharded")
    shutil.copytree(hf_dir, d)
    sd = load_file(os.path.join(d, "model.safetensors"))
    os.remove(os.path.join(d, "model.safetensors"))
    keys = sorted(sd)
    half = len(keys) // 2
    save_file({k: sd[k] for k in keys[:half]},
              os.path.join(d, "model-00001-of-00002.safetensors"))
    save_file({k: sd[k] for k in keys[half:]},
              os.path.join(d, "model-00002-of-00002.safetensors"))
    wm = {k: ("model-00001-of-00002.safetensors" if i < half else
              "model-00002-of-00002.safetensors")
          for i, k in enumerate(keys)}
    with open(os.path.join(d, "model.safetensors.index.json"), "w") as f:
        json.dump({"weight_map": wm}, f)

    cfg = load_hf_config(d)
    ours = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    load_hf_weights(ours, d)
    ref = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    load_hf_weights(ref, hf_dir)
    for (n1, p1), (_, p2) in zip(ours.named_parameters(),
                                 ref.named_parameters()):
        assert torch.equal(p1, p2), n1


def test_tokenizer_roundtrip_and_specials(hf_dir):
    tok = load_tokenizer(hf_dir)
    assert isinstance(tok, HFTokenizer)
    ids = tok.encode("the quick brown fox", add_special_tokens=True)
    assert ids[0] == tok.bos_token_id
    text = tok.decode(ids)
    assert "quick" in text and "fox" in text
    assert tok.vocab_size > 0


def test_train_on_hf_dir_end_to_end(hf_dir, tmp_path):
    """The trainer accepts model_name_or_path = a local HF dir: loads
    real weights + tokenizer, trains LoRA, saves an adapter the engine
    can serve on those weights."""
    from datatunerx_amd.serve.engine import InferenceEngine, build_model
    from datatunerx_amd.train.run import main as train_main
    out = str(tmp_path / "out")
    train_main([
        "--model_name_or_path", hf_dir, "--output_dir", out,
        "--max_steps", "2", "--per_device_train_batch_size", "2",
        "--synthetic_examples", "8", "--block_size", "32",
        "--lora_rank", "4", "--lora_target", "q_proj,v_proj",
        "--logging_steps", "0"])
    ckpt = os.path.join(out, "checkpoint")
    assert os.path.exists(os.path.join(ckpt, "adapter_model.safetensors"))
    model = build_model(hf_dir, torch.device("cpu"), adapter_dir=ckpt)
    eng = InferenceEngine(model, tokenizer=load_tokenizer(hf_dir),
                          template="vanilla", device=torch.device("cpu"))
    txt = eng.chat([{"role": "user", "content": "hello"}], max_tokens=4)
    assert isinstance(txt, str)


def test_hf_export_roundtrip(hf_dir, tmp_path):
    """save_hf_model output loads in transformers with matching logits
    (full-param checkpoint interchange)."""
    from datatunerx_amd.models.hf_io import save_hf_model
    cfg = load_hf_config(hf_dir)
    ours = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    load_hf_weights(ours, hf_dir)
    out = str(tmp_path / "export")
    save_hf_model(ours, out)
    ref = transformers.LlamaForCausalLM.from_pretrained(
        out, torch_dtype=torch.float32, attn_implementation="eager")
    ref.eval()
    ours.eval()
    ids = torch.randint(0, 320, (1, 13))
    with torch.no_grad():
        assert torch.allclose(ours(ids), ref(ids).logits, atol=2e-4,
                              rtol=1e-3)


def test_list_eos_and_engine_stop_set(hf_dir, tmp_path):
    """llama3-style checkpoints declare eos_token_id as a LIST; the
    tokenizer keeps all of them as stop ids and the engine stops on any
    (a single-eos engine would run past <|eot_id|> on real llama3)."""
    import shutil
    d = str(tmp_path / "l3style")
    shutil.copytree(hf_dir, d)
    cfgp = os.path.join(d, "config.json")
    with open(cfgp) as f:
        hc = json.load(f)
    hc["eos_token_id"] = [2, 7, 9]
    with open(cfgp, "w") as f:
        json.dump(hc, f)
    tok = HFTokenizer.from_dir(d)
    assert tok.eos_token_id == 2
    assert tok.stop_token_ids == {2, 7, 9}
    # decode drops every stop id, not just the primary eos
    ids = tok.encode("fox", add_special_tokens=False)
    assert tok.decode(ids + [7, 9, 2]) == tok.decode(ids)

    from datatunerx_amd.models import LlamaConfig
    from datatunerx_amd.serve.engine import InferenceEngine
    model = LlamaForCausalLM(LlamaConfig.tiny(), dtype=torch.float32)
    eng = InferenceEngine(model, tokenizer=tok, template="llama3",
                          device=torch.device("cpu"))
    assert {2, 7, 9} <= eng._stop_ids
    assert eng._is_stop(7) and not eng._is_stop(5)


def test_llama3_template_encode():
    from datatunerx_amd.data.dataset import ByteTokenizer
    from datatunerx_amd.data.templates import get_template
    t = get_template("llama3")
    tok = ByteTokenizer()
    src, tgt = t.encode_oneturn(tok, "hi", "yo", None, "sys")
    text = bytes(i - 3 for i in src if i >= 3).decode()
    assert "<|start_header_id|>system<|end_header_id|>\n\nsys<|eot_id|>" \
        in text
    assert text.endswith("assistant<|end_header_id|>\n\n")
    assert "<|eot_id|>" in t.stop_words


def test_qwen2_checkpoint_matches_transformers_logits(tmp_path):
    """Qwen2-family (q/k/v bias, GQA, tied head) loads through the same
    HF-dir path and matches transformers' fp32 logits."""
    d = str(tmp_path / "qwen2")
    cfg = transformers.Qwen2Config(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, max_position_embeddings=128,
        rms_norm_eps=1e-5, rope_theta=10000.0,
        tie_word_embeddings=True, attn_implementation="eager")
    torch.manual_seed(11)
    hf = transformers.Qwen2ForCausalLM(cfg)
    hf.save_pretrained(d, safe_serialization=True)

    mycfg = load_hf_config(d)
    assert mycfg.attention_bias and mycfg.num_key_value_heads == 2
    model = LlamaForCausalLM(mycfg, lora=False, dtype=torch.float32)
    n = load_hf_weights(model, d)
    assert n >= 2 * 9 + 3          # per-layer tensors + embed/norm/head
    # bias actually landed (Qwen2 inits bias to zeros=False? randn no —
    # from_config inits bias zero; force nonzero to prove the add runs)
    with torch.no_grad():
        for lyr, mylyr in zip(hf.model.layers, model.layers):
            for p in ("q", "k", "v"):
                b = getattr(lyr.self_attn, f"{p}_proj").bias
                b.copy_(torch.randn_like(b) * 0.5)
                getattr(mylyr.self_attn, f"{p}_bias").copy_(b)
    ids = torch.randint(0, 256, (2, 17))
    with torch.no_grad():
        ref = hf(ids).logits
        got = model(ids)
    assert torch.allclose(ref, got, atol=2e-4, rtol=1e-3), \
        (ref - got).abs().max()


def test_qwen2_export_roundtrip(tmp_path):
    """save_hf_model writes qkv bias back under HF names; reloading
    reproduces identical logits."""
    from datatunerx_amd.models import LlamaConfig
    from datatunerx_amd.models.hf_io import save_hf_model
    cfg = LlamaConfig.tiny(attention_bias=True)
    torch.manual_seed(4)
    m = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    m.init_random(seed=9)
    with torch.no_grad():
        for lyr in m.layers:
            for p in ("q", "k", "v"):
                b = getattr(lyr.self_attn, f"{p}_bias")
                b.copy_(torch.randn_like(b) * 0.3)
    d = str(tmp_path / "exp")
    save_hf_model(m, d)
    m2 = LlamaForCausalLM(load_hf_config(d), lora=False,
                          dtype=torch.float32)
    load_hf_weights(m2, d)
    ids = torch.randint(0, cfg.vocab_size, (1, 23))
    with torch.no_grad():
        assert torch.equal(m(ids), m2(ids))


def test_qwen2_lora_train_and_serve(tmp_path):
    """LoRA fine-tune + engine serve on a Qwen2 checkpoint dir (bias
    params stay frozen under LoRA; adapter applies on top)."""
    from datatunerx_amd.serve.engine import InferenceEngine, build_model
    from datatunerx_amd.train.run import main as train_main
    d = str(tmp_path / "qwen2")
    cfg = transformers.Qwen2Config(
        vocab_size=256, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=2,
        num_key_value_heads=2, max_position_embeddings=256,
        tie_word_embeddings=False, attn_implementation="eager")
    transformers.Qwen2ForCausalLM(cfg).save_pretrained(
        d, safe_serialization=True)
    out = str(tmp_path / "out")
    train_main([
        "--model_name_or_path", d, "--output_dir", out,
        "--max_steps", "2", "--per_device_train_batch_size", "2",
        "--synthetic_examples", "8", "--block_size", "32",
        "--lora_rank", "4", "--logging_steps", "0"])
    ckpt = os.path.join(out, "checkpoint")
    model = build_model(d, torch.device("cpu"), adapter_dir=ckpt)
    assert model.layers[0].self_attn.q_bias is not None
    eng = InferenceEngine(model, template="chatml",
                          device=torch.device("cpu"))
    assert isinstance(eng.chat([{"role": "user", "content": "hi"}],
                               max_tokens=4), str)


def test_engine_rejects_overlong_prompt():
    from datatunerx_amd.models import LlamaConfig
    from datatunerx_amd.serve.engine import InferenceEngine
    m = LlamaForCausalLM(LlamaConfig.tiny(), lora=False,
                         dtype=torch.float32)
    eng = InferenceEngine(m, template="vanilla",
                          device=torch.device("cpu"))
    with pytest.raises(ValueError, match="exceeds the model context"):
        eng.generate(list(range(1, 300)), max_new_tokens=2)


def test_qwen3_checkpoint_matches_transformers_logits(tmp_path):
    """Qwen3-family (per-head q/k RMSNorm before RoPE, decoupled
    head_dim, no bias) matches transformers' fp32 logits."""
    d = str(tmp_path / "qwen3")
    cfg = transformers.Qwen3Config(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=32,     # 4*32 != 64: decoupled
        max_position_embeddings=128, rms_norm_eps=1e-5,
        rope_theta=10000.0, tie_word_embeddings=False,
        attn_implementation="eager")
    torch.manual_seed(13)
    hf = transformers.Qwen3ForCausalLM(cfg)
    with torch.no_grad():                       # nontrivial norm weights
        for lyr in hf.model.layers:
            lyr.self_attn.q_norm.weight.uniform_(0.5, 1.5)
            lyr.self_attn.k_norm.weight.uniform_(0.5, 1.5)
    hf.save_pretrained(d, safe_serialization=True)

    mycfg = load_hf_config(d)
    assert mycfg.qk_norm and not mycfg.attention_bias
    assert mycfg.head_dim == 32
    model = LlamaForCausalLM(mycfg, lora=False, dtype=torch.float32)
    load_hf_weights(model, d)
    ids = torch.randint(0, 256, (2, 19))
    with torch.no_grad():
        ref = hf(ids).logits
        got = model(ids)
    assert torch.allclose(ref, got, atol=2e-4, rtol=1e-3), \
        (ref - got).abs().max()


def test_qwen3_export_reloads_in_transformers(tmp_path):
    from datatunerx_amd.models import LlamaConfig
    from datatunerx_amd.models.hf_io import save_hf_model
    cfg = LlamaConfig.tiny(qk_norm=True)
    torch.manual_seed(6)
    m = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    m.init_random(seed=2)
    with torch.no_grad():
        for lyr in m.layers:
            lyr.self_attn.q_norm.uniform_(0.5, 1.5)
            lyr.self_attn.k_norm.uniform_(0.5, 1.5)
    d = str(tmp_path / "exp3")
    save_hf_model(m, d)
    with open(os.path.join(d, "config.json")) as f:
        assert json.load(f)["model_type"] == "qwen3"
    ref = transformers.AutoModelForCausalLM.from_pretrained(
        d, torch_dtype=torch.float32, attn_implementation="eager")
    ref.eval()
    ids = torch.randint(0, cfg.vocab_size, (1, 21))
    with torch.no_grad():
        assert torch.allclose(m(ids), ref(ids).logits, atol=2e-4,
                              rtol=1e-3)


def test_sentencepiece_tokenizer_backend(tmp_path):
    """tokenizer.model (SentencePiece) backend: same interface as the
    tokenizers backend — encode/decode roundtrip, special ids, and the
    trainer/engine-facing add_special_tokens contract."""
    spm = pytest.importorskip("sentencepiece")
    d = str(tmp_path / "spdir")
    os.makedirs(d)
    corpus = os.path.join(d, "c.txt")
    with open(corpus, "w") as f:
        f.write("the quick brown fox\nllama fine tune\nhello world\n" * 40)
    spm.SentencePieceTrainer.train(
        input=corpus, model_prefix=os.path.join(d, "tokenizer"),
        vocab_size=80, model_type="bpe", minloglevel=2)
    tok = load_tokenizer(d)
    assert isinstance(tok, HFTokenizer) and tok.kind == "sp"
    assert tok.bos_token_id == 1 and tok.eos_token_id == 2
    ids = tok.encode("quick fox", add_special_tokens=True)
    assert ids[0] == tok.bos_token_id and len(ids) > 1
    text = tok.decode(ids)
    assert "quick" in text and "fox" in text
    assert tok.vocab_size == 80
    # dataset pipeline consumes it like any tokenizer
    from datatunerx_amd.data.dataset import SFTDataset
    ds = SFTDataset.from_rows(
        [{"instruction": "quick fox", "response": "hello world"}],
        tok, template_name="default", cutoff_len=64)
    assert len(ds) == 1 and any(l != -100 for l in ds[0]["labels"])


def test_malformed_config_clear_error(tmp_path):
    d = str(tmp_path / "bad")
    os.makedirs(d)
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump({"architectures": ["LlamaForCausalLM"],
                   "vocab_size": 128}, f)
    with pytest.raises(ValueError, match="missing required"):
        load_hf_config(d)
    with open(os.path.join(d, "config.json"), "w") as f:
        json.dump({"architectures": ["FalconForCausalLM"]}, f)
    with pytest.raises(ValueError, match="unsupported architecture"):
        load_hf_config(d)
