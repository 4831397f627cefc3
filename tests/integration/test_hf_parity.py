"""Interop proof (scaffold/convert.py): weights converted from
`transformers` models produce the same logits/loss as transformers
itself — an external ground truth for the architecture (RoPE layout,
GQA packing, Conv1D transposes, GELU flavor), not a self-consistency
check."""

import pytest
import torch
import torch.nn.functional as F

from scaffold.config import load_config
from scaffold.convert import load_hf
from scaffold.models import build_model

transformers = pytest.importorskip("transformers")


def _our_loss(model, tok, tgt):
    model.eval()
    with torch.no_grad():
        return model(tok, tgt).float()


def _ref_loss(logits, tgt):
    return F.cross_entropy(logits.view(-1, logits.shape[-1]).float(),
                           tgt.view(-1))


def test_gpt2_matches_transformers():
    from transformers import GPT2Config, GPT2LMHeadModel

    torch.manual_seed(17)
    hf = GPT2LMHeadModel(GPT2Config(
        vocab_size=128, n_positions=32, n_embd=64, n_layer=2, n_head=4,
        activation_function="gelu_new",   # == our tanh-approx GELU
        resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
    )).eval()

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=32",
    ])
    ours = build_model(cfg)
    load_hf(ours, hf.state_dict())

    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    got = _our_loss(ours, tok, tgt)
    assert torch.allclose(got, ref, rtol=1e-4, atol=1e-5), (got, ref)


def test_llama_matches_transformers():
    from transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(23)
    hf = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, rope_theta=500000.0,
        rms_norm_eps=1e-5, attention_bias=False, mlp_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
    )).eval()

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=32",
        "model.rope_theta=500000.0",
    ])
    ours = build_model(cfg)
    load_hf(ours, hf.state_dict())

    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    got = _our_loss(ours, tok, tgt)
    assert torch.allclose(got, ref, rtol=1e-4, atol=1e-5), (got, ref)


def test_llama_generate_matches_transformers_greedy():
    """End-to-end decode interop: our KV-cached Generator reproduces
    transformers' greedy continuation on converted weights."""
    from transformers import LlamaConfig, LlamaForCausalLM

    from scaffold.generate import Generator

    torch.manual_seed(29)
    hf = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=500000.0,
        rms_norm_eps=1e-5, attention_bias=False, mlp_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
    )).eval()
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
        "model.rope_theta=500000.0",
    ])
    ours = build_model(cfg).eval()
    load_hf(ours, hf.state_dict())

    prompt = torch.randint(0, 128, (2, 8))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=12, do_sample=False,
                          pad_token_id=0)
        got = Generator(ours).generate(prompt, max_new_tokens=12)
    assert torch.equal(got, ref)


def test_load_hf_rejects_shape_mismatch():
    from transformers import GPT2Config, GPT2LMHeadModel

    hf = GPT2LMHeadModel(GPT2Config(
        vocab_size=128, n_positions=32, n_embd=64, n_layer=2, n_head=4)).eval()
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=96", "model.vocab_size=128", "model.max_seq_len=32",
    ])
    ours = build_model(cfg)
    with pytest.raises(ValueError, match="shape"):
        load_hf(ours, hf.state_dict())


def test_roundtrip_export_llama():
    """ours -> HF layout -> transformers forward == ours; and
    re-importing the export reproduces the original weights exactly."""
    from transformers import LlamaConfig, LlamaForCausalLM

    from scaffold.convert import to_hf_llama

    torch.manual_seed(31)
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=32",
        "model.rope_theta=500000.0",
    ])
    ours = build_model(cfg).eval()
    exported = to_hf_llama(ours)

    hf = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=32, rope_theta=500000.0,
        rms_norm_eps=1e-5, attention_bias=False, mlp_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
    )).eval()
    missing, unexpected = hf.load_state_dict(exported, strict=False)
    assert not unexpected, unexpected
    assert all("rotary_emb" in m or "bias" not in m for m in missing), missing

    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    assert torch.allclose(_our_loss(ours, tok, tgt), ref,
                          rtol=1e-4, atol=1e-5)

    # import(export(x)) == x bitwise
    clone = build_model(cfg)
    load_hf(clone, exported)
    for (n, a), (_, b) in zip(ours.named_parameters(),
                              clone.named_parameters()):
        assert torch.equal(a, b), n


def test_roundtrip_export_gpt2():
    from transformers import GPT2Config, GPT2LMHeadModel

    from scaffold.convert import to_hf_gpt2

    torch.manual_seed(37)
    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=32",
    ])
    ours = build_model(cfg).eval()
    exported = to_hf_gpt2(ours)
    hf = GPT2LMHeadModel(GPT2Config(
        vocab_size=128, n_positions=32, n_embd=64, n_layer=2, n_head=4,
        activation_function="gelu_new",
        resid_pdrop=0.0, embd_pdrop=0.0, attn_pdrop=0.0,
    )).eval()
    missing, unexpected = hf.load_state_dict(exported, strict=False)
    assert not unexpected, unexpected
    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    assert torch.allclose(_our_loss(ours, tok, tgt), ref,
                          rtol=1e-4, atol=1e-5)


def test_save_load_model_single_file(tmp_path):
    """Single-file weight export round-trips, including tied tensors."""
    from scaffold.convert import load_model, save_model

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=32",
    ])
    torch.manual_seed(61)
    a = build_model(cfg).eval()
    p = str(tmp_path / "m.safetensors")
    save_model(a, p)
    torch.manual_seed(62)
    b = build_model(cfg).eval()
    load_model(b, p)
    for (n, x), (_, y) in zip(a.state_dict().items(), b.state_dict().items()):
        assert torch.equal(x, y), n
    # tying preserved: loading into b kept lm_head sharing wte storage
    assert b.lm_head.weight.data_ptr() == b.wte.weight.data_ptr()


def test_llama31_rope_scaling_matches_transformers():
    """Llama-3.1-style rope scaling: our wavelength-dependent rule ==
    transformers' (loss parity on converted weights with scaling ON)."""
    from transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(101)
    hf = LlamaForCausalLM(LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, rope_theta=500000.0,
        rms_norm_eps=1e-5, attention_bias=False, mlp_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
        rope_scaling={"rope_type": "llama3", "factor": 8.0,
                      "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                      "original_max_position_embeddings": 16},
    )).eval()
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
        "model.rope_theta=500000.0", "model.rope_scaling=llama3",
        "model.rope_factor=8.0", "model.rope_low_freq_factor=1.0",
        "model.rope_high_freq_factor=4.0", "model.rope_orig_max_pos=16",
    ])
    ours = build_model(cfg)
    load_hf(ours, hf.state_dict())
    tok = torch.randint(0, 128, (2, 48))     # beyond orig_max_pos=16
    tgt = torch.randint(0, 128, (2, 48))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    got = _our_loss(ours, tok, tgt)
    assert torch.allclose(got, ref, rtol=1e-4, atol=1e-5), (got, ref)


def test_linear_rope_scaling_changes_tables():
    from scaffold import ops

    c0, s0 = ops.rope_tables(16, 32, 1e4, "cpu")
    c1, s1 = ops.rope_tables(16, 32, 1e4, "cpu", scaling="linear",
                             factor=2.0)
    assert not torch.allclose(c0, c1)
    # position 2k with factor 2 == position k unscaled
    assert torch.allclose(c1[10], c0[5], atol=1e-6)
    assert torch.allclose(s1[10], s0[5], atol=1e-6)


def test_mixtral_matches_transformers():
    """Mixtral interop: converted weights + our top-2 SwiGLU-expert
    routing reproduce transformers' MixtralForCausalLM loss."""
    from transformers import MixtralConfig, MixtralForCausalLM

    torch.manual_seed(141)
    hf = MixtralForCausalLM(MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=32, rope_theta=1e6,
        rms_norm_eps=1e-5, attention_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
        router_jitter_noise=0.0,
    )).eval()
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=96",
        "model.vocab_size=128", "model.max_seq_len=32",
        "model.rope_theta=1e6", "model.n_experts=4", "model.moe_top_k=2",
    ])
    ours = build_model(cfg)
    load_hf(ours, hf.state_dict())
    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    got = _our_loss(ours, tok, tgt)
    assert torch.allclose(got, ref, rtol=1e-4, atol=1e-5), (got, ref)


def test_mixtral_generate_matches_transformers_greedy():
    from transformers import MixtralConfig, MixtralForCausalLM

    from scaffold.generate import Generator

    torch.manual_seed(151)
    hf = MixtralForCausalLM(MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=64, rope_theta=1e6,
        rms_norm_eps=1e-5, attention_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
        router_jitter_noise=0.0,
    )).eval()
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=96",
        "model.vocab_size=128", "model.max_seq_len=64",
        "model.rope_theta=1e6", "model.n_experts=4", "model.moe_top_k=2",
    ])
    ours = build_model(cfg).eval()
    load_hf(ours, hf.state_dict())
    prompt = torch.randint(0, 128, (2, 8))
    with torch.no_grad():
        ref = hf.generate(prompt, max_new_tokens=10, do_sample=False,
                          pad_token_id=0)
        got = Generator(ours).generate(prompt, max_new_tokens=10)
    assert torch.equal(got, ref)


def test_roundtrip_export_mixtral():
    from transformers import MixtralConfig, MixtralForCausalLM

    from scaffold.convert import to_hf_mixtral

    torch.manual_seed(161)
    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=96",
        "model.vocab_size=128", "model.max_seq_len=32",
        "model.rope_theta=1e6", "model.n_experts=4", "model.moe_top_k=2",
    ])
    ours = build_model(cfg).eval()
    exported = to_hf_mixtral(ours)
    hf = MixtralForCausalLM(MixtralConfig(
        vocab_size=128, hidden_size=64, intermediate_size=96,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        max_position_embeddings=32, rope_theta=1e6,
        rms_norm_eps=1e-5, attention_bias=False,
        tie_word_embeddings=False, attn_implementation="eager",
        router_jitter_noise=0.0,
    )).eval()
    missing, unexpected = hf.load_state_dict(exported, strict=False)
    assert not unexpected, unexpected
    tok = torch.randint(0, 128, (2, 32))
    tgt = torch.randint(0, 128, (2, 32))
    with torch.no_grad():
        ref = _ref_loss(hf(tok).logits, tgt)
    assert torch.allclose(_our_loss(ours, tok, tgt), ref,
                          rtol=1e-4, atol=1e-5)
    # import(export(x)) == x
    clone = build_model(cfg)
    load_hf(clone, exported)
    for (n, a), (_, b) in zip(ours.named_parameters(),
                              clone.named_parameters()):
        assert torch.equal(a, b), n
