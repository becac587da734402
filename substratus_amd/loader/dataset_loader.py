"""Dataset loader — replaces `substratusai/images//dataset-loader-http`
(reference examples/datasets/k8s-instructions.yaml:5-10).

PARAM_URLS: space/comma-separated URLs downloaded into
/content/artifacts/*.jsonl (http(s) via httpx when the network exists;
file:// always). PARAM_SYNTHETIC_ROWS: generate a synthetic instruction
dataset offline. Writes completed.json when done.
"""

from __future__ import annotations

import json
import os
import re
import shutil
import sys

from .model_loader import load_params


def run(artifacts_dir: str = "/content/artifacts",
        params: dict | None = None) -> str:
    params = params if params is not None else load_params()
    os.makedirs(artifacts_dir, exist_ok=True)
    sentinel = os.path.join(artifacts_dir, "completed.json")
    if os.path.exists(sentinel):
        print("completed.json present; skipping")
        return artifacts_dir
    urls = re.split(r"[,\s]+", params.get("urls", "").strip())
    urls = [u for u in urls if u]
    n = 0
    for url in urls:
        n += _fetch(url, artifacts_dir)
    rows = int(params.get("synthetic_rows", 0) or 0)
    if rows:
        path = os.path.join(artifacts_dir, "synthetic.jsonl")
        with open(path, "w") as f:
            for i in range(rows):
                f.write(json.dumps({
                    "prompt": f"Instruction {i}: describe resource {i % 97}.",
                    "completion": f"Resource {i % 97} is item {i}. " * 8,
                }) + "\n")
        n += 1
    if n == 0:
        raise SystemExit("nothing to load: set PARAM_URLS or "
                         "PARAM_SYNTHETIC_ROWS")
    with open(sentinel, "w") as f:
        json.dump({"completed": True, "files": n}, f)
    return artifacts_dir


def _fetch(url: str, out: str) -> int:
    name = url.rsplit("/", 1)[-1] or "data.jsonl"
    dst = os.path.join(out, name)
    if url.startswith("file://"):
        shutil.copyfile(url[len("file://"):], dst)
        return 1
    try:
        import httpx
        r = httpx.get(url, follow_redirects=True, timeout=60)
        r.raise_for_status()
        with open(dst, "wb") as f:
            f.write(r.content)
        return 1
    except Exception as e:
        print(f"fetch {url} failed: {e}")
        raise


if __name__ == "__main__":
    run(sys.argv[1] if len(sys.argv) > 1 else "/content/artifacts")
