"""KV-cached decoding equivalence: greedy generation with the cache
matches recompute-from-scratch generation (the model's training forward
repurposed for logits), for both LM families (fp32, CPU)."""

import torch
import torch.nn.functional as F

from scaffold.config import load_config
from scaffold.generate import Generator
from scaffold.models import build_model


def _tiny(name, extra=()):
    cfg = load_config(None, [
        f"model.name={name}", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=64",
        "dtype=fp32", "device=cpu", *extra,
    ])
    torch.manual_seed(9)
    return build_model(cfg)


def _ref_greedy(gen: Generator, prompt, n):
    """no-cache reference: full forward from scratch every step."""
    from scaffold.generate import KVCache

    m = gen.m
    attn0 = m.blocks[0].attn
    n_kv = attn0.n_kv if gen.is_llama else attn0.n_heads
    dtype = next(m.parameters()).dtype
    out = prompt
    for _ in range(n):
        cache = KVCache(len(m.blocks), prompt.shape[0], n_kv,
                        out.shape[1], attn0.head_dim, prompt.device, dtype)
        logits = gen._forward_tokens(out, cache, 0)
        out = torch.cat([out, logits.argmax(-1, keepdim=True)], dim=1)
    return out


def test_gpt2_greedy_cache_matches_recompute():
    m = _tiny("gpt2")
    g = Generator(m)
    prompt = torch.randint(0, 128, (2, 7))
    got = g.generate(prompt, max_new_tokens=8)
    want = _ref_greedy(g, prompt, 8)
    assert torch.equal(got, want)


def test_llama_greedy_cache_matches_recompute():
    m = _tiny("llama", ["model.n_kv_heads=2", "model.d_ff=128"])
    g = Generator(m)
    prompt = torch.randint(0, 128, (2, 5))
    got = g.generate(prompt, max_new_tokens=8)
    want = _ref_greedy(g, prompt, 8)
    assert torch.equal(got, want)


def test_decode_logits_match_training_forward():
    """the decode path's block math == the training path's block math:
    per-position logits from Generator._forward_tokens equal the
    training forward's logits (via its loss on a delta probe... here
    directly: rebuild logits with the model's own modules)."""
    m = _tiny("gpt2")
    g = Generator(m)
    tokens = torch.randint(0, 128, (1, 9))
    from scaffold.generate import KVCache

    cache = KVCache(len(m.blocks), 1, m.blocks[0].attn.n_heads, 9,
                    m.blocks[0].attn.head_dim, tokens.device,
                    next(m.parameters()).dtype)
    logits_dec = g._forward_tokens(tokens, cache, 0)
    # training-path logits for the LAST position: run model internals
    pos = torch.arange(9)
    x = m.wte(tokens) + m.wpe(pos)
    for blk in m.blocks:
        x = blk(x)
    from scaffold import ops

    x = ops.layer_norm(x, m.ln_f.weight, m.ln_f.bias, m.ln_f.eps)
    logits_train = F.linear(x[:, -1], m.lm_head.weight)
    assert torch.allclose(logits_dec, logits_train, atol=1e-5)


def test_static_decode_matches_dynamic():
    """generate_graphed's static-shape step (padded cache, device-side
    position) == the dynamic-shape path, both families (CPU eager)."""
    for fam, extra in [("gpt2", []),
                       ("llama", ["model.n_kv_heads=2", "model.d_ff=128"])]:
        m = _tiny(fam, extra) if extra else _tiny(fam)
        g = Generator(m)
        torch.manual_seed(4)
        prompt = torch.randint(0, 128, (2, 7))
        a = g.generate(prompt, max_new_tokens=9)
        b = g.generate_graphed(prompt, max_new_tokens=9, use_graph=False)
        assert torch.equal(a, b), fam


def test_topk_topp_sampling():
    """top_k=1 == greedy for any temperature; top_k/top_p masks keep
    samples inside the allowed sets."""
    import torch

    from scaffold.generate import Generator

    logits = torch.tensor([[4.0, 3.0, 2.0, 1.0, 0.0],
                           [0.0, 1.0, 2.0, 3.0, 4.0]])
    g = torch.Generator().manual_seed(0)
    # top_k=1 at temperature>0 is greedy
    nxt = Generator._sample(logits, 1.0, 1, 0.0, g)
    assert nxt[:, 0].tolist() == [0, 4]
    # top_k=2: only the two best ids can ever be drawn
    seen = set()
    for _ in range(64):
        nxt = Generator._sample(logits, 2.0, 2, 0.0, g)
        seen.update((int(nxt[0, 0]), int(nxt[1, 0]) - 3 + 10))
    assert seen <= {0, 1, 10, 11}, seen
    # top_p tiny: nucleus collapses to the argmax
    nxt = Generator._sample(logits, 5.0, 0, 1e-6, g)
    assert nxt[:, 0].tolist() == [0, 4]
    # greedy path (temperature 0) ignores top_k/top_p
    nxt = Generator._sample(logits, 0.0, 3, 0.5, None)
    assert nxt[:, 0].tolist() == [0, 4]


def test_generate_with_sampling_runs():
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=1", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=32",
    ])
    torch.manual_seed(1)
    m = build_model(cfg).eval()
    g = torch.Generator().manual_seed(7)
    out = Generator(m).generate(torch.randint(0, 64, (2, 4)), 8,
                                temperature=0.8, top_k=10, top_p=0.9,
                                generator=g)
    assert out.shape == (2, 12)
    assert out.max() < 64 and out.min() >= 0


def test_eos_early_stop_and_freeze():
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=64",
    ])
    torch.manual_seed(47)
    m = build_model(cfg).eval()
    g = Generator(m)
    prompt = torch.tensor([[1, 2, 3], [4, 5, 6]])
    ref = g.generate(prompt, 10)
    # use each row's FIRST greedy continuation as EOS: both rows must
    # freeze immediately and the tail repeat the eos id
    eos = int(ref[0, 3])
    out = g.generate(prompt, 10, eos_token_id=eos)
    # wherever a row hits eos, everything after is eos
    for b in range(2):
        toks = out[b, 3:].tolist()
        if eos in toks:
            i = toks.index(eos)
            assert all(t == eos for t in toks[i:]), toks
    # rows identical to the unconstrained rollout up to their first eos
    for b in range(2):
        toks = out[b, 3:].tolist()
        reft = ref[b, 3:].tolist()
        n = toks.index(eos) if eos in toks else len(toks)
        assert toks[:n] == reft[:n]


def test_speculative_matches_greedy_exactly():
    """Speculative decode == plain greedy regardless of draft quality:
    same-family draft, cross-quality draft, and k sweep."""
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    def make(seed, layers=2, d=64):
        cfg = load_config(None, [
            "model.name=llama", f"model.n_layers={layers}",
            "model.n_heads=4", "model.n_kv_heads=2", f"model.d_model={d}",
            f"model.d_ff={2 * d}", "model.vocab_size=128",
            "model.max_seq_len=96",
        ])
        torch.manual_seed(seed)
        return build_model(cfg).eval()

    target = Generator(make(51, layers=2, d=64))
    prompt = torch.randint(0, 128, (1, 6))
    ref = target.generate(prompt, 20)

    # a GOOD draft (the target itself) and a BAD draft (random tiny)
    for draft_m, tag in ((make(51, 2, 64), "self"), (make(99, 1, 32), "rand")):
        for k in (1, 3, 5):
            stats = {}
            out = target.generate_speculative(prompt, 20,
                                              Generator(draft_m), k=k,
                                              stats=stats)
            assert torch.equal(out, ref), (tag, k)
            assert stats["rounds"] >= 1
    # the self-draft must accept nearly everything
    stats = {}
    target.generate_speculative(prompt, 20, Generator(make(51, 2, 64)),
                                k=4, stats=stats)
    assert stats["accepted"] >= stats["rounds"] * 3, stats


def test_speculative_gpt2_family():
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=4",
        "model.d_model=64", "model.vocab_size=128", "model.max_seq_len=96",
    ])
    torch.manual_seed(53)
    target = Generator(build_model(cfg).eval())
    torch.manual_seed(77)
    cfg2 = load_config(None, [
        "model.name=gpt2", "model.n_layers=1", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=128", "model.max_seq_len=96",
    ])
    draft = Generator(build_model(cfg2).eval())
    prompt = torch.randint(0, 128, (1, 5))
    ref = target.generate(prompt, 16)
    out = target.generate_speculative(prompt, 16, draft, k=4)
    assert torch.equal(out, ref)


def test_speculative_eos_matches_generate():
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=96",
    ])
    torch.manual_seed(51)
    target = Generator(build_model(cfg).eval())
    prompt = torch.randint(0, 128, (1, 6))
    ref_free = target.generate(prompt, 20)
    eos = int(ref_free[0, 6 + 4])         # 5th generated token as eos
    ref = target.generate(prompt, 20, eos_token_id=eos)
    torch.manual_seed(99)
    cfg2 = load_config(None, [
        "model.name=llama", "model.n_layers=1", "model.n_heads=2",
        "model.n_kv_heads=1", "model.d_model=32", "model.d_ff=64",
        "model.vocab_size=128", "model.max_seq_len=96",
    ])
    draft = Generator(build_model(cfg2).eval())
    out = target.generate_speculative(prompt, 20, draft, k=3,
                                      eos_token_id=eos)
    assert torch.equal(out, ref)


def test_speculative_randomized_sweep():
    """Deterministic randomized stress: many (prompt_len, k, max_new)
    combinations against plain greedy — guards the cache-rollback
    arithmetic across round boundaries and truncation edges."""
    import random

    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    cfg = load_config(None, [
        "model.name=llama", "model.n_layers=2", "model.n_heads=4",
        "model.n_kv_heads=2", "model.d_model=64", "model.d_ff=128",
        "model.vocab_size=128", "model.max_seq_len=128",
    ])
    torch.manual_seed(57)
    target = Generator(build_model(cfg).eval())
    torch.manual_seed(58)
    cfg_d = load_config(None, [
        "model.name=llama", "model.n_layers=1", "model.n_heads=2",
        "model.n_kv_heads=1", "model.d_model=32", "model.d_ff=64",
        "model.vocab_size=128", "model.max_seq_len=128",
    ])
    draft = Generator(build_model(cfg_d).eval())

    rng = random.Random(0)
    for _ in range(12):
        s0 = rng.randint(1, 12)
        n = rng.randint(1, 25)
        k = rng.randint(1, 6)
        prompt = torch.randint(0, 128, (1, s0))
        ref = target.generate(prompt, n)
        out = target.generate_speculative(prompt, n, draft, k=k)
        assert torch.equal(out, ref), (s0, n, k)


def test_moe_model_decodes():
    """MoE GPT decodes through all three paths: generate (KV cache) ==
    recompute-from-scratch, and the continuous batcher agrees."""
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator, KVCache
    from scaffold.models import build_model
    from scaffold.serve import ContinuousBatcher

    cfg = load_config(None, [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=48",
        "model.n_experts=4",
    ])
    torch.manual_seed(121)
    m = build_model(cfg).eval()
    g = Generator(m)
    prompt = torch.randint(0, 64, (1, 5))
    out = g.generate(prompt, 12)

    # recompute-from-scratch each step (no incremental cache reuse)
    toks = prompt.clone()
    with torch.no_grad():
        for _ in range(12):
            cache = KVCache(2, 1, 2, toks.shape[1], 16, toks.device,
                            torch.float32)
            logits = g._forward_tokens(toks, cache, 0)
            toks = torch.cat([toks, logits.argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, toks)

    cb = ContinuousBatcher(m, max_slots=2)
    try:
        r = cb.submit(prompt[0].tolist(), 12)
        assert r.event.wait(60) and r.error is None
        assert r.result == out[0, 5:].tolist()
    finally:
        cb.close()


def test_speculative_on_moe_model():
    import torch

    from scaffold.config import load_config
    from scaffold.generate import Generator
    from scaffold.models import build_model

    over = [
        "model.name=gpt2", "model.n_layers=2", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=64",
        "model.n_experts=4",
    ]
    torch.manual_seed(131)
    target = Generator(build_model(load_config(None, over)).eval())
    torch.manual_seed(132)
    draft = Generator(build_model(load_config(None, [
        "model.name=gpt2", "model.n_layers=1", "model.n_heads=2",
        "model.d_model=32", "model.vocab_size=64", "model.max_seq_len=64",
    ])).eval())
    prompt = torch.randint(0, 64, (1, 5))
    ref = target.generate(prompt, 14)
    out = target.generate_speculative(prompt, 14, draft, k=3)
    assert torch.equal(out, ref)
