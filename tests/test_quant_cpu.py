"""GGUF format + quantization CPU tests (config 4 of BASELINE: the
llama2-13b-chat-gguf quantized serving path, reference
examples/llama2-13b-chat-gguf/)."""

import os

import pytest
import torch

from substratus_amd.models import CausalLM, get_config
from substratus_amd.models.gguf import (dequantize_q4_0, dequantize_q8_0,
                                        export_model_gguf, quantize_q4_0,
                                        quantize_q8_0, read_gguf, write_gguf,
                                        GGML_F32)
from substratus_amd.models.quant import (QuantLinear, load_gguf_into_model,
                                         quantize_model)


def test_q8_0_roundtrip_error():
    torch.manual_seed(0)
    w = torch.randn(8, 256)
    packed = quantize_q8_0(w)
    back = dequantize_q8_0(packed, 256)
    err = (w - back).abs().max() / w.abs().max()
    assert err < 0.01  # int8: ~0.4% of blockmax


def test_q4_0_roundtrip_error():
    torch.manual_seed(1)
    w = torch.randn(8, 256)
    packed = quantize_q4_0(w)
    back = dequantize_q4_0(packed, 256)
    err = (w - back).abs().max() / w.abs().max()
    assert err < 0.15  # 4-bit: ~1/16 of blockmax


def test_gguf_write_read_roundtrip(tmp_path):
    torch.manual_seed(2)
    path = str(tmp_path / "t.gguf")
    w1 = torch.randn(4, 64)
    w2 = torch.randn(32)
    write_gguf(path, {
        "a.weight": (quantize_q8_0(w1), 8, [4, 64]),
        "b.weight": (w2, GGML_F32, [32]),
    }, metadata={"general.name": "test", "n": 7})
    meta, tensors = read_gguf(path)
    assert meta["general.name"] == "test"
    assert meta["n"] == 7
    torch.testing.assert_close(tensors["b.weight"].data, w2)
    torch.testing.assert_close(tensors["a.weight"].dequantize(),
                               dequantize_q8_0(quantize_q8_0(w1), 64))


@pytest.mark.parametrize("quant", ["q4_0", "q8_0"])
def test_quant_linear_matches_dequant_ref(quant):
    torch.manual_seed(3)
    from substratus_amd.parallel.tp import Linear
    lin = Linear(128, 96, bias=False, dtype=torch.float32)
    lin.weight.data.normal_(0, 0.1)
    ql = QuantLinear.from_linear(lin, quant)
    x = torch.randn(5, 128)
    want = torch.nn.functional.linear(x, ql._dequant_cpu())
    got = ql(x)
    torch.testing.assert_close(got, want, atol=1e-4, rtol=1e-3)


def test_quantized_engine_generates():
    from substratus_amd.serve.engine import Engine, EngineConfig
    torch.manual_seed(4)
    cfg = EngineConfig(model="llama-tiny", dtype=torch.float32, max_batch=2,
                       max_seq_len=64, kv_blocks=16, quant="q8_0")
    eng = Engine(cfg, device="cpu")
    from substratus_amd.models.quant import QuantLinear as QL
    n_quant = sum(1 for m in eng.model.modules() if isinstance(m, QL))
    assert n_quant >= 4 * eng.mcfg.num_layers  # all projections swapped
    sid = eng.add_request([1, 2, 3, 4], max_new_tokens=3)
    toks = []
    while eng.has_work():
        toks.extend(o.token for o in eng.step())
    assert len(toks) == 3


def test_gguf_export_load_roundtrip(tmp_path):
    """Loader's GGUF export → server's GGUF load: same generation output
    as quantizing the model in memory."""
    torch.manual_seed(5)
    mcfg = get_config("llama-tiny")
    model = CausalLM(mcfg, device="cpu", dtype=torch.float32)
    model.init_random(seed=13)
    path = str(tmp_path / "m.gguf")
    export_model_gguf(path, model.state_dict(), "llama-tiny", quant="q8_0")
    assert os.path.getsize(path) > 0
    # load into a fresh model
    model2 = CausalLM(mcfg, device="cpu", dtype=torch.float32)
    quant = load_gguf_into_model(model2, path, "cpu", torch.float32)
    assert quant == "q8_0"
    # reference: quantize the original in memory
    model_q = quantize_model(model, "q8_0")
    x = torch.randint(0, 512, (1, 12))
    with torch.no_grad():
        a = model_q.forward_train(x)
        b = model2.forward_train(x)
    torch.testing.assert_close(a, b, atol=1e-3, rtol=1e-3)


def test_loader_gguf_format(tmp_path):
    from substratus_amd.loader.model_loader import run
    out = run(str(tmp_path), params={"name": "llama-tiny", "format": "gguf",
                                     "quant": "q8_0"})
    files = os.listdir(out)
    assert any(f.endswith(".gguf") for f in files)


def test_gguf_reader_rejects_corrupt_files(tmp_path):
    """Truncated or non-GGUF files raise cleanly (ValueError/struct.error),
    never hang or return garbage tensors."""
    import struct
    import pytest
    from substratus_amd.models import gguf

    bad_magic = tmp_path / "bad.gguf"
    bad_magic.write_bytes(b"NOTG" + b"\x00" * 40)
    with pytest.raises(ValueError, match="not a GGUF"):
        gguf.read_gguf(str(bad_magic))

    truncated = tmp_path / "trunc.gguf"
    truncated.write_bytes(struct.pack("<IIQQ", gguf.GGUF_MAGIC,
                                      gguf.GGUF_VERSION, 5, 5))
    with pytest.raises((ValueError, struct.error, EOFError)):
        gguf.read_gguf(str(truncated))


def test_llamacpp_named_gguf_loads(tmp_path):
    """An ecosystem GGUF with llama.cpp tensor names (blk.N.attn_q etc. +
    llama.cpp's q/k rope row-permutation, as TheBloke-style files carry —
    the reference's llama2-13b-chat-gguf example) loads into the native
    model and generates the SAME tokens as loading the native-named export
    of the same weights."""
    import torch
    from substratus_amd.models import gguf
    from substratus_amd.models.config import get_config
    from substratus_amd.models.model import CausalLM
    from substratus_amd.models.quant import load_gguf_into_model
    from substratus_amd.serve.engine import Engine, EngineConfig

    cfg = get_config("llama-tiny")
    torch.manual_seed(4)
    ref = CausalLM(cfg, device="meta", dtype=torch.float32)
    ref = ref.to_empty(device="cpu")
    ref.init_random(4)
    sd = {k: v.float() for k, v in ref.state_dict().items()}

    hq, hkv, d = cfg.num_heads, cfg.num_kv_heads, cfg.head_dim

    def perm_fwd(w, n_head):
        # llama.cpp convert.py: within each head rows reorder via
        # reshape(n_head, 2, d/2, k).swapaxes(1, 2)
        n, k = w.shape
        return (w.reshape(n_head, 2, d // 2, k).swapaxes(1, 2)
                 .reshape(n, k))

    tensors = {}
    def q40(name, w, logical_shape=None):
        tensors[name] = (gguf.quantize_q4_0(w), gguf.GGML_Q4_0,
                         logical_shape or list(w.shape))
    def f32(name, w):
        tensors[name] = (w.contiguous(), gguf.GGML_F32, list(w.shape))

    f32("token_embd.weight", sd["embed_tokens.weight"])
    f32("output_norm.weight", sd["final_norm.weight"])
    q40("output.weight", sd["lm_head.weight"])
    for i in range(cfg.num_layers):
        p = f"blocks.{i}."
        f32(f"blk.{i}.attn_norm.weight", sd[p + "norm1.weight"])
        f32(f"blk.{i}.ffn_norm.weight", sd[p + "norm2.weight"])
        wqkv = sd[p + "wqkv.weight"]
        wq, wk, wv = wqkv.split([hq * d, hkv * d, hkv * d], dim=0)
        q40(f"blk.{i}.attn_q.weight", perm_fwd(wq, hq))
        q40(f"blk.{i}.attn_k.weight", perm_fwd(wk, hkv))
        q40(f"blk.{i}.attn_v.weight", wv)
        q40(f"blk.{i}.attn_output.weight", sd[p + "wo.weight"])
        gu = sd[p + "w_gateup.weight"]
        g, u = gu.chunk(2, dim=0)
        q40(f"blk.{i}.ffn_gate.weight", g)
        q40(f"blk.{i}.ffn_up.weight", u)
        q40(f"blk.{i}.ffn_down.weight", sd[p + "w_down.weight"])
    path = str(tmp_path / "llamacpp.gguf")
    gguf.write_gguf(path, tensors, {"general.architecture": "llama"})

    def gen(gguf_path):
        eng = Engine(EngineConfig(model="llama-tiny", dtype=torch.float32,
                                  max_batch=1, max_seq_len=64, kv_blocks=16,
                                  seed=5))
        load_gguf_into_model(eng.model, gguf_path, eng.device, torch.float32)
        eng.add_request([5, 9, 2, 7], max_new_tokens=6)
        toks = []
        while eng.has_work():
            toks += [o.token for o in eng.step()]
        return toks

    # native-named export of the SAME weights for comparison
    native = {}
    for k, v in sd.items():
        if v.dim() == 2 and "embed" not in k:
            native[k] = (gguf.quantize_q4_0(v), gguf.GGML_Q4_0,
                         list(v.shape))
        else:
            native[k] = (v.contiguous(), gguf.GGML_F32, list(v.shape))
    npath = str(tmp_path / "native.gguf")
    gguf.write_gguf(npath, native, {"substratus.quant": "q4_0"})

    assert gen(path) == gen(npath)


def test_gguf_nondefault_alignment_roundtrip(tmp_path):
    """Ecosystem GGUF files may set general.alignment != 32; the reader must
    honor it for both the data-section base and per-tensor offsets."""
    import torch
    from substratus_amd.models.gguf import (GGML_F32, read_gguf, write_gguf)
    a = torch.randn(7, 32)
    b = torch.randn(3, 64)
    path = str(tmp_path / "aligned.gguf")
    write_gguf(path, {"a": (a, GGML_F32, list(a.shape)),
                      "b": (b, GGML_F32, list(b.shape))},
               metadata={"x": "y"}, alignment=64)
    meta, tensors = read_gguf(path)
    assert meta["general.alignment"] == 64
    assert torch.equal(tensors["a"].dequantize(), a)
    assert torch.equal(tensors["b"].dequantize(), b)


def _kq_model():
    import torch
    from substratus_amd.models.config import ModelConfig, register
    from substratus_amd.models.model import CausalLM
    cfg = register(ModelConfig(
        name="ktest-tiny", family="llama", hidden_size=256,
        intermediate_size=512, num_layers=2, num_heads=4, num_kv_heads=2,
        vocab_size=512, max_position_embeddings=256))
    torch.manual_seed(11)
    m = CausalLM(cfg, device="meta", dtype=torch.float32)
    m = m.to_empty(device="cpu")
    m.init_random(7)
    return cfg, m


def test_kquant_codec_error_bounds():
    """q4_K beats q4_0 (sub-block mins) and q6_K beats both."""
    import torch
    from substratus_amd.models.gguf import QUANT_CODECS
    torch.manual_seed(5)
    w = torch.randn(16, 1024) * 0.04
    rms = {}
    for q in ("q4_0", "q4_K", "q6_K"):
        _, _, qf, df = QUANT_CODECS[q]
        back = df(qf(w), 1024)
        rms[q] = float((back - w).pow(2).mean().sqrt() /
                       w.pow(2).mean().sqrt())
    assert rms["q4_K"] < rms["q4_0"]
    assert rms["q6_K"] < rms["q4_K"] / 2
    assert rms["q6_K"] < 0.03


def test_q4k_ecosystem_file_generates_and_matches_dequant(tmp_path):
    """A llama.cpp-named q4_K file (ecosystem Q4_K class) loads through
    the K-quant path and generates EXACTLY the tokens of a model whose
    weights are the codec's dequantized values (the QuantLinear CPU path
    is F.linear over the same dequant)."""
    import torch
    from substratus_amd.models import gguf
    from substratus_amd.models.model import CausalLM
    from substratus_amd.models.quant import QuantLinear, load_gguf_into_model
    from substratus_amd.serve.engine import Engine, EngineConfig

    cfg, ref = _kq_model()
    sd = {k: v.float() for k, v in ref.state_dict().items()}
    hq, hkv, d = cfg.num_heads, cfg.num_kv_heads, cfg.head_dim

    def perm_fwd(w, n_head):
        n, k = w.shape
        return (w.reshape(n_head, 2, d // 2, k).swapaxes(1, 2)
                 .reshape(n, k))

    tensors = {}

    def q4k(name, w):
        tensors[name] = (gguf.quantize_q4_K(w), gguf.GGML_Q4_K,
                         list(w.shape))

    def f32(name, w):
        tensors[name] = (w.contiguous(), gguf.GGML_F32, list(w.shape))

    f32("token_embd.weight", sd["embed_tokens.weight"])
    f32("output_norm.weight", sd["final_norm.weight"])
    q4k("output.weight", sd["lm_head.weight"])
    for i in range(cfg.num_layers):
        p = f"blocks.{i}."
        f32(f"blk.{i}.attn_norm.weight", sd[p + "norm1.weight"])
        f32(f"blk.{i}.ffn_norm.weight", sd[p + "norm2.weight"])
        wq, wk, wv = sd[p + "wqkv.weight"].split(
            [hq * d, hkv * d, hkv * d], dim=0)
        q4k(f"blk.{i}.attn_q.weight", perm_fwd(wq, hq))
        q4k(f"blk.{i}.attn_k.weight", perm_fwd(wk, hkv))
        q4k(f"blk.{i}.attn_v.weight", wv)
        q4k(f"blk.{i}.attn_output.weight", sd[p + "wo.weight"])
        g, u = sd[p + "w_gateup.weight"].chunk(2, dim=0)
        q4k(f"blk.{i}.ffn_gate.weight", g)
        q4k(f"blk.{i}.ffn_up.weight", u)
        q4k(f"blk.{i}.ffn_down.weight", sd[p + "w_down.weight"])
    path = str(tmp_path / "ktest.Q4_K.gguf")
    gguf.write_gguf(path, tensors)

    qm = CausalLM(cfg, device="meta", dtype=torch.float32)
    qm = qm.to_empty(device="cpu")
    qm.init_random(7)
    quant = load_gguf_into_model(qm, path, "cpu", torch.float32)
    assert quant == "q4_K"
    assert isinstance(qm.blocks[0].wqkv, QuantLinear)
    assert qm.blocks[0].wqkv.quant == "q4_K"

    # reference: plain model with the dequantized weights
    dm = CausalLM(cfg, device="meta", dtype=torch.float32)
    dm = dm.to_empty(device="cpu")
    dm.init_random(7)
    dsd = dm.state_dict()
    for name, mod in qm.named_modules():
        if isinstance(mod, QuantLinear):
            dsd[name + ".weight"] = mod._dequant_cpu()
    dm.load_state_dict(dsd)

    toks = {}
    for tag, model in (("quant", qm), ("dequant", dm)):
        eng = Engine(EngineConfig(model="ktest-tiny", dtype=torch.float32,
                                  max_batch=1, max_seq_len=64,
                                  kv_blocks=16), device="cpu", model=model)
        eng.add_request([3, 1, 4, 1, 5], max_new_tokens=8)
        out = []
        while eng.has_work():
            out.extend(o.token for o in eng.step())
        toks[tag] = out
    assert toks["quant"] == toks["dequant"], toks


def test_mixed_kquant_fusion_loads(tmp_path):
    """Q4_K_M-style mixing: attn_v is q6_K while attn_q/k are q4_K — the
    fused wqkv re-encodes at q6_K and the weights stay within codec
    error of the originals."""
    import torch
    from substratus_amd.models import gguf
    from substratus_amd.models.model import CausalLM
    from substratus_amd.models.quant import QuantLinear, load_gguf_into_model

    cfg, ref = _kq_model()
    sd = {k: v.float() for k, v in ref.state_dict().items()}
    hq, hkv, d = cfg.num_heads, cfg.num_kv_heads, cfg.head_dim

    def perm_fwd(w, n_head):
        n, k = w.shape
        return (w.reshape(n_head, 2, d // 2, k).swapaxes(1, 2)
                 .reshape(n, k))

    tensors = {"token_embd.weight": (sd["embed_tokens.weight"].contiguous(),
                                     gguf.GGML_F32, [cfg.vocab_size, 256]),
               "output_norm.weight": (sd["final_norm.weight"].contiguous(),
                                      gguf.GGML_F32, [256]),
               "output.weight": (gguf.quantize_q6_K(sd["lm_head.weight"]),
                                 gguf.GGML_Q6_K,
                                 list(sd["lm_head.weight"].shape))}
    for i in range(cfg.num_layers):
        p = f"blocks.{i}."
        tensors[f"blk.{i}.attn_norm.weight"] = (
            sd[p + "norm1.weight"].contiguous(), gguf.GGML_F32, [256])
        tensors[f"blk.{i}.ffn_norm.weight"] = (
            sd[p + "norm2.weight"].contiguous(), gguf.GGML_F32, [256])
        wq, wk, wv = sd[p + "wqkv.weight"].split(
            [hq * d, hkv * d, hkv * d], dim=0)
        tensors[f"blk.{i}.attn_q.weight"] = (
            gguf.quantize_q4_K(perm_fwd(wq, hq)), gguf.GGML_Q4_K,
            list(wq.shape))
        tensors[f"blk.{i}.attn_k.weight"] = (
            gguf.quantize_q4_K(perm_fwd(wk, hkv)), gguf.GGML_Q4_K,
            list(wk.shape))
        tensors[f"blk.{i}.attn_v.weight"] = (
            gguf.quantize_q6_K(wv), gguf.GGML_Q6_K, list(wv.shape))
        tensors[f"blk.{i}.attn_output.weight"] = (
            gguf.quantize_q4_K(sd[p + "wo.weight"]), gguf.GGML_Q4_K,
            list(sd[p + "wo.weight"].shape))
        g, u = sd[p + "w_gateup.weight"].chunk(2, dim=0)
        tensors[f"blk.{i}.ffn_gate.weight"] = (
            gguf.quantize_q4_K(g), gguf.GGML_Q4_K, list(g.shape))
        tensors[f"blk.{i}.ffn_up.weight"] = (
            gguf.quantize_q4_K(u), gguf.GGML_Q4_K, list(u.shape))
        tensors[f"blk.{i}.ffn_down.weight"] = (
            gguf.quantize_q6_K(sd[p + "w_down.weight"]), gguf.GGML_Q6_K,
            list(sd[p + "w_down.weight"].shape))
    path = str(tmp_path / "ktest.Q4_K_M.gguf")
    gguf.write_gguf(path, tensors)

    qm = CausalLM(cfg, device="meta", dtype=torch.float32)
    qm = qm.to_empty(device="cpu")
    qm.init_random(7)
    load_gguf_into_model(qm, path, "cpu", torch.float32)
    wqkv = qm.blocks[0].wqkv
    assert isinstance(wqkv, QuantLinear) and wqkv.quant == "q6_K"
    back = wqkv._dequant_cpu()
    orig = sd["blocks.0.wqkv.weight"]
    rel = float((back - orig).pow(2).mean().sqrt() /
                orig.pow(2).mean().sqrt())
    assert rel < 0.12, rel
    # down stayed pure q6_K (no re-encode)
    assert qm.blocks[0].w_down.quant == "q6_K"
