"""Training data: jsonl files from /content/data (the Dataset contract —
reference examples/datasets/k8s-instructions.yaml writes *.jsonl) and a
synthetic generator for benchmarking (BASELINE: synthetic data, random-init
weights).

No network ⇒ no real tokenizer downloads: jsonl text is byte-level tokenized
(UTF-8 bytes offset into the model vocab), which exercises the identical
compute path.
"""

from __future__ import annotations

import json
import os
from typing import Iterator, Optional

import torch


def byte_tokenize(text: str, vocab_size: int) -> list[int]:
    return [2 + (b % (vocab_size - 3)) for b in text.encode("utf-8")]


class JsonlDataset:
    """Reads {"prompt","completion"} or {"text"} lines from *.jsonl files,
    applying an optional prompt_template (reference param: prompt_template —
    examples/falcon-7b-instruct/finetuned-model.yaml)."""

    def __init__(self, data_dir: str, vocab_size: int, seq_len: int,
                 prompt_template: str = "{prompt}\n{completion}"):
        self.files = sorted(
            os.path.join(data_dir, f) for f in os.listdir(data_dir)
            if f.endswith(".jsonl"))
        if not self.files:
            raise FileNotFoundError(f"no *.jsonl files in {data_dir}")
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        self.template = prompt_template
        self.examples: list[list[int]] = []
        for path in self.files:
            with open(path) as f:
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    row = json.loads(line)
                    if "text" in row:
                        text = row["text"]
                    else:
                        text = self.template.format(
                            prompt=row.get("prompt", ""),
                            completion=row.get("completion", ""))
                    self.examples.append(byte_tokenize(text, vocab_size))

    def __len__(self) -> int:
        return len(self.examples)

    def batches(self, micro_batch: int, dp_rank: int = 0,
                dp_size: int = 1, seed: int = 0) -> Iterator[torch.Tensor]:
        """Packed [micro_batch, seq_len+1] batches, sharded across DP ranks."""
        g = torch.Generator().manual_seed(seed)
        order = torch.randperm(len(self.examples), generator=g).tolist()
        order = order[dp_rank::dp_size]
        stream: list[int] = []
        chunk = self.seq_len + 1
        rows: list[list[int]] = []
        for idx in order:
            stream.extend(self.examples[idx] + [0])
            while len(stream) >= chunk:
                rows.append(stream[:chunk])
                stream = stream[chunk:]
                if len(rows) == micro_batch:
                    yield torch.tensor(rows, dtype=torch.int64)
                    rows = []


class SyntheticDataset:
    def __init__(self, vocab_size: int, seq_len: int, seed: int = 0):
        self.vocab_size = vocab_size
        self.seq_len = seq_len
        self.seed = seed

    def batches(self, micro_batch: int, dp_rank: int = 0,
                dp_size: int = 1) -> Iterator[torch.Tensor]:
        g = torch.Generator().manual_seed(self.seed + dp_rank)
        while True:
            yield torch.randint(0, self.vocab_size,
                                (micro_batch, self.seq_len + 1), generator=g)
