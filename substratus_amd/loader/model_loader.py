"""Model loader — replaces the reference's external
`substratusai/model-loader-huggingface` image.

Contract (reference examples/llama2-7b/base-model.yaml:8-13,
examples/llama2-13b-chat-gguf/base-model.yaml:7-9): reads PARAM_NAME (model
id), optional PARAM_FILES filter; HUGGING_FACE_HUB_TOKEN via secret-ref env;
writes weights into /content/artifacts/ and a completed.json sentinel; skips
when completed.json already exists (docs/design.md "Reconcile Logic").

Offline path (this environment has no egress): PARAM_NAME naming a known
architecture produces deterministic random-init safetensors of that
architecture. Online path: huggingface_hub snapshot_download when the hub is
reachable. GGUF files are passed through for the quantized serve path.
"""

from __future__ import annotations

import json
import os
import sys

import torch


def load_params(path: str = "/content/params.json") -> dict:
    params = {}
    if os.path.exists(path):
        with open(path) as f:
            params.update(json.load(f))
    for k, v in os.environ.items():
        if k.startswith("PARAM_"):
            params[k[len("PARAM_"):].lower()] = v
    return params


def run(artifacts_dir: str = "/content/artifacts",
        params: dict | None = None) -> str:
    params = params if params is not None else load_params()
    os.makedirs(artifacts_dir, exist_ok=True)
    sentinel = os.path.join(artifacts_dir, "completed.json")
    if os.path.exists(sentinel):
        print("completed.json present; skipping (resume semantics)")
        return artifacts_dir
    name = params.get("name")
    if not name:
        raise SystemExit("PARAM_NAME is required (model id)")
    files_filter = params.get("files", "")

    if _try_hub_download(name, files_filter, artifacts_dir):
        pass
    elif str(params.get("format", "")).lower() == "gguf" or \
            ".gguf" in files_filter:
        _write_random_gguf(name, artifacts_dir, params)
    else:
        _write_random_init(name, artifacts_dir, params)

    # total artifact byte size: the controller propagates this into
    # Model.status.artifacts.sizeBytes and the Server controller's HBM
    # admission (gpu.count × 288 GB) reads it — without it admission can
    # never trigger
    size = _artifacts_size(artifacts_dir)
    with open(sentinel, "w") as f:
        json.dump({"completed": True, "name": name, "sizeBytes": size}, f)
    print(f"loaded {name} into {artifacts_dir} ({size} bytes)")
    return artifacts_dir


def _artifacts_size(artifacts_dir: str) -> int:
    total = 0
    for root, _dirs, files in os.walk(artifacts_dir):
        for fn in files:
            if fn == "completed.json":
                continue
            total += os.path.getsize(os.path.join(root, fn))
    return total


def _try_hub_download(name: str, files_filter: str, out: str) -> bool:
    if "/" not in name:
        return False  # config names like llama2-7b are always offline
    try:
        from huggingface_hub import snapshot_download
        kwargs = {"repo_id": name, "local_dir": out}
        if files_filter:
            kwargs["allow_patterns"] = files_filter.split(",")
        snapshot_download(**kwargs)
        return True
    except Exception as e:  # offline / no token
        print(f"hub download unavailable ({type(e).__name__}); "
              f"falling back to random-init weights")
        return False


def _write_random_init(name: str, out: str, params: dict) -> None:
    from safetensors.torch import save_file
    from ..models.config import get_config
    from ..models.model import CausalLM
    cfg = get_config(name)
    dtype = torch.bfloat16 if params.get("dtype", "bf16") == "bf16" \
        else torch.float32
    model = CausalLM(cfg, device="meta", dtype=dtype)
    model = model.to_empty(device="cpu")
    model.init_random(seed=int(params.get("seed", 0)))
    tensors = {k: v.detach().contiguous()
               for k, v in model.state_dict().items()}
    save_file(tensors, os.path.join(out, "model.safetensors"))
    # real tokenizer artifact (serve/tokenizer.py loads it; byte-level BPE
    # since vocabularies aren't downloadable offline)
    from ..serve.tokenizer import write_byte_level_tokenizer
    write_byte_level_tokenizer(os.path.join(out, "tokenizer.json"))
    with open(os.path.join(out, "config.json"), "w") as f:
        json.dump({"model": cfg.name, "dtype": str(dtype).split(".")[-1],
                   "source": "random-init"}, f)


def _write_random_gguf(name: str, out: str, params: dict) -> None:
    """GGUF export of a random-init model (offline stand-in for the
    reference's llama2-13b-chat-gguf example)."""
    from ..models.config import get_config
    from ..models.gguf import export_model_gguf
    from ..models.model import CausalLM
    cfg = get_config(name)
    model = CausalLM(cfg, device="meta", dtype=torch.float32)
    model = model.to_empty(device="cpu")
    model.init_random(seed=int(params.get("seed", 0)))
    quant = params.get("quant", "q4_0")
    export_model_gguf(os.path.join(out, f"{cfg.name}.{quant}.gguf"),
                      model.state_dict(), cfg.name, quant)
    with open(os.path.join(out, "config.json"), "w") as f:
        json.dump({"model": cfg.name, "format": "gguf", "quant": quant,
                   "source": "random-init"}, f)


if __name__ == "__main__":
    run(sys.argv[1] if len(sys.argv) > 1 else "/content/artifacts")
