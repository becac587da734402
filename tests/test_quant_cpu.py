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
