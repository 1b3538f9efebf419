"""Text dataloaders: tokenized corpus -> fixed-length context windows.

Capability parity with the reference's WikiText pipeline
(``examples/wikitext103/dataloaders/dataloaders.py:22-84``): a raw text
file is tokenized once, concatenated, chunked into ``context_length``
windows, and cached as ``.npz`` so later runs lazy-load
(dataloaders.py:70-84's cache contract).  The collate returns
``(batch, batch.clone())`` like the reference (dataloaders.py:22-24).

Tokenization: a HuggingFace tokenizer when ``tokenizer_json`` is given
(the offline `tokenizers` wheel — no network), else a byte-level fallback
(vocab 256) so the pipeline works in fully offline environments.
"""

from __future__ import annotations

import os
from typing import Callable, Optional

import numpy as np
import torch


def _tokenize(text: str, tokenizer_json: Optional[str]) -> np.ndarray:
    if tokenizer_json:
        from tokenizers import Tokenizer

        tok = Tokenizer.from_file(tokenizer_json)
        return np.asarray(tok.encode(text).ids, dtype=np.int64)
    return np.frombuffer(text.encode("utf-8"), dtype=np.uint8).astype(np.int64)


def load_text_dataset(
    path: str,
    context_length: int = 512,
    tokenizer_json: Optional[str] = None,
    cache_dir: Optional[str] = None,
) -> torch.Tensor:
    """Returns [n_windows, context_length] int64 token windows, cached."""
    cache_dir = cache_dir or os.path.dirname(os.path.abspath(path))
    cache = os.path.join(
        cache_dir,
        f"{os.path.basename(path)}.ctx{context_length}.npz",
    )
    if os.path.isfile(cache):
        ids = np.load(cache)["ids"]
    else:
        with open(path, "r", errors="ignore") as f:
            ids = _tokenize(f.read(), tokenizer_json)
        np.savez_compressed(cache, ids=ids)
    n = len(ids) // context_length
    return torch.from_numpy(
        np.ascontiguousarray(ids[: n * context_length]).reshape(n, context_length)
    )


def _collate(batch):
    x = torch.stack(batch)
    return x, x.clone()


def make_text_dataloader(
    path: str,
    batch_size: int = 8,
    context_length: int = 512,
    tokenizer_json: Optional[str] = None,
) -> Callable:
    """A Task-compatible dataloader factory over a local text file."""

    def get_dataloader():
        windows = load_text_dataset(path, context_length, tokenizer_json)
        return torch.utils.data.DataLoader(
            windows, batch_size=batch_size, shuffle=False, collate_fn=_collate
        )

    return get_dataloader
