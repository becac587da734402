"""nbwatch — file-change watcher for notebook sync
(reference containertools/cmd/nbwatch/main.go:30-105, 105 LoC Go).

Watches /content (non-recursive, plus one level of non-dot subdirectories,
skipping data/, model/, artifacts/ — the bucket mounts) and emits
{"index": N, "path": ..., "op": CREATE|WRITE|REMOVE} JSON lines on stdout.
Poll-based mtime scanning (no fsnotify dependency); the CLI consumes the
stream to copy changed files back from the notebook pod
(reference internal/client/sync.go:184-273).
"""

from __future__ import annotations

import json
import os
import sys
import time

SKIP_DIRS = {"data", "model", "artifacts"}


def watched_files(root: str) -> dict[str, float]:
    out: dict[str, float] = {}
    try:
        entries = os.listdir(root)
    except FileNotFoundError:
        return out
    for e in entries:
        if e.startswith("."):
            continue
        full = os.path.join(root, e)
        if os.path.isfile(full):
            out[full] = os.path.getmtime(full)
        elif os.path.isdir(full) and e not in SKIP_DIRS:
            for sub in os.listdir(full):
                if sub.startswith("."):
                    continue
                sf = os.path.join(full, sub)
                if os.path.isfile(sf):
                    out[sf] = os.path.getmtime(sf)
    return out


def watch(root: str = "/content", interval: float = 0.5,
          iterations: int = -1) -> None:
    index = 0
    prev = watched_files(root)
    n = 0
    while iterations < 0 or n < iterations:
        time.sleep(interval)
        n += 1
        cur = watched_files(root)
        for path, mtime in cur.items():
            if path not in prev:
                op = "CREATE"
            elif mtime != prev[path]:
                op = "WRITE"
            else:
                continue
            print(json.dumps({"index": index, "path": path, "op": op}),
                  flush=True)
            index += 1
        for path in prev:
            if path not in cur:
                print(json.dumps({"index": index, "path": path,
                                  "op": "REMOVE"}), flush=True)
                index += 1
        prev = cur


if __name__ == "__main__":
    watch(sys.argv[1] if len(sys.argv) > 1 else "/content")
