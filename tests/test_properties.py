"""Property-based tests (hypothesis) for the pure-math/pure-data layers:
quantization codec error bounds, strategic-merge semantics, tokenizer
roundtrips. These run on random inputs every time — the fixed-seed unit
tests elsewhere pin exact values."""


import torch
from hypothesis import given, settings, strategies as st

from substratus_amd.kube.memapi import deep_merge
from substratus_amd.models import gguf
from substratus_amd.serve import tokenizer as tok


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 8), st.integers(1, 8), st.random_module())
def test_q8_0_roundtrip_bound(rows, blocks, _rng):
    w = torch.randn(rows, blocks * 32)
    packed = gguf.quantize_q8_0(w)
    back = gguf.dequantize_q8_0(packed, w.shape[-1])
    # int8 symmetric quant: |err| <= scale/2 = amax/254 per block, plus the
    # fp16 rounding of the stored scale (~2^-11 relative)
    amax = w.reshape(-1, 32).abs().amax(dim=1, keepdim=True)
    bound = (amax / 254.0 + amax * 2e-3 + 1e-6).expand(-1, 32).reshape(w.shape)
    assert ((back - w).abs() <= bound).all()


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 8), st.integers(1, 8), st.random_module())
def test_q4_0_roundtrip_bound(rows, blocks, _rng):
    w = torch.randn(rows, blocks * 32)
    packed = gguf.quantize_q4_0(w)
    back = gguf.dequantize_q4_0(packed, w.shape[-1])
    # llama.cpp q4_0 uses the SIGNED max: d = max/-8, values clamp to
    # [-8, 7], so the +amax side can err by up to amax/8 (plus fp16 scale
    # rounding)
    amax = w.reshape(-1, 32).abs().amax(dim=1, keepdim=True)
    bound = (amax / 8.0 + amax * 3e-3 + 1e-6).expand(-1, 32).reshape(w.shape)
    assert ((back - w).abs() <= bound).all()


json_scalars = st.one_of(st.integers(-10, 10), st.text(max_size=5),
                         st.booleans())
json_vals = st.recursive(
    json_scalars,
    lambda c: st.dictionaries(st.text(max_size=4), c, max_size=3),
    max_leaves=10)


@settings(max_examples=50, deadline=None)
@given(st.dictionaries(st.text(max_size=4), json_vals, max_size=4))
def test_deep_merge_identity_and_idempotent(d):
    # merging a dict into itself is the dict; merging {} changes nothing
    assert deep_merge(d, {}) == d
    assert deep_merge(d, d) == d
    assert deep_merge({}, d) == d


@settings(max_examples=50, deadline=None)
@given(st.text(max_size=64))
def test_byte_tokenizer_roundtrip(text):
    t = tok.ByteTokenizer(32000)
    out = t.decode(t.encode(text))
    assert out == text


def _hf_tok():
    global _HF_TOK_CACHE
    try:
        return _HF_TOK_CACHE
    except NameError:
        import os
        import tempfile
        d = tempfile.mkdtemp()
        p = os.path.join(d, "tokenizer.json")
        tok.write_byte_level_tokenizer(p)
        _HF_TOK_CACHE = tok.HFTokenizer(p)
        return _HF_TOK_CACHE


@settings(max_examples=20, deadline=None)
@given(st.text(st.characters(blacklist_categories=("Cs",)), max_size=48))
def test_hf_byte_level_tokenizer_roundtrip(text):
    t = _hf_tok()
    assert t.decode(t.encode(text)) == text


@given(st.integers(0, 2**32 - 1),
       st.sampled_from(["q4_K", "q6_K"]),
       st.integers(1, 4))
@settings(max_examples=20, deadline=None)
def test_kquant_codec_roundtrip_bounds(seed, quant, rows):
    """K-quant codecs: dequant(quant(w)) stays within the codec's step
    bound for every sub-block (q4_K: d*sc covers [min,max] in 15 steps;
    q6_K: 31 steps of d*sc per 16-group)."""
    import torch
    from substratus_amd.models.gguf import QUANT_CODECS
    g = torch.Generator().manual_seed(seed)
    w = torch.randn(rows, 512, generator=g) * 0.1
    _, _, qf, df = QUANT_CODECS[quant]
    back = df(qf(w), 512)
    if quant == "q4_K":
        grp = w.reshape(-1, 32)
        span = (grp.amax(1) - grp.amin(1)).clamp(min=1e-6)
        step = span / 15.0
        err = (back.reshape(-1, 32) - grp).abs().amax(1)
        # quantization error <= one step (+ 6-bit scale/min rounding slack)
        assert bool((err <= step * 1.2 + 1e-3).all())
    else:
        grp = w.reshape(-1, 16)
        step = grp.abs().amax(1).clamp(min=1e-6) / 31.0
        err = (back.reshape(-1, 16) - grp).abs().amax(1)
        assert bool((err <= step * 1.2 + 1e-3).all())


@given(st.integers(0, 2**32 - 1), st.sampled_from(["q4_K", "q6_K"]))
@settings(max_examples=10, deadline=None)
def test_kquant_repack_involution(seed, quant):
    """repack_k (SoA for the GPU kernels) and unrepack_k are exact
    inverses on random valid blocks."""
    import torch
    from substratus_amd.models.gguf import QUANT_CODECS
    from substratus_amd.models.quant import repack_k, unrepack_k
    g = torch.Generator().manual_seed(seed)
    w = torch.randn(4, 768, generator=g)
    packed = QUANT_CODECS[quant][2](w)
    s, qs = repack_k(packed, 768, quant)
    assert torch.equal(unrepack_k(s, qs, 768, quant), packed)


@given(st.text(min_size=0, max_size=200))
@settings(max_examples=60, deadline=None)
def test_stream_decoder_exact_for_any_text(text):
    """StreamDecoder pushed one token at a time reconstructs ANY unicode
    text exactly (byte-level tokenizer splits multi-byte chars across
    tokens freely), and no intermediate delta ever contains U+FFFD."""
    from hypothesis import assume
    from substratus_amd.serve.tokenizer import ByteTokenizer, StreamDecoder
    assume("�" not in text)   # a literal U+FFFD in the INPUT is emitted as-is
    tok = ByteTokenizer(32000)
    ids = tok.encode(text, add_bos=False)
    dec = StreamDecoder(tok)
    parts = [dec.push(t) for t in ids]
    parts.append(dec.flush())
    assert "".join(parts) == text
    assert all("�" not in p for p in parts)
