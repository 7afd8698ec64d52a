"""Tokenizer loading for converted checkpoints.

The reference relies on `transformers.AutoTokenizer` against the HF hub;
this environment has no egress, so tokenizer FILES (tokenizer.json /
tokenizer.model + configs) must already sit in the checkpoint directory —
`convert_hf_checkpoint` copies them into the converted layout so one
directory serves weights AND tokenizer (ref server/from_pretrained.py +
client tokenizer usage in benchmarks/benchmark_inference.py).
"""
from __future__ import annotations

import os
from typing import Optional


def load_tokenizer(path: str, **kw):
    """AutoTokenizer over a local directory (converted layout or original HF
    checkout). Returns None-raising errors early with a clear message."""
    from transformers import AutoTokenizer

    has_files = any(
        os.path.exists(os.path.join(path, f))
        for f in ("tokenizer.json", "tokenizer.model", "vocab.json"))
    if not has_files:
        raise FileNotFoundError(
            f"no tokenizer files under {path!r} (expected tokenizer.json / "
            "tokenizer.model; convert_hf_checkpoint copies them)")
    return AutoTokenizer.from_pretrained(path, local_files_only=True, **kw)
