"""Byte-level tokenizer for the front door.

The reference proxies prompt text straight to vLLM, which owns tokenization.
This framework serves random-init models on synthetic workloads (no network,
no checkpoint vocabularies), so the front door uses a deterministic byte-level
codec: UTF-8 byte ``b`` maps to token id ``b + OFFSET``.  It round-trips any
string, needs no assets, and keeps ids well inside every model's vocab.
Requests may also bypass it by passing ``prompt`` as a list of token ids
(mirrors vLLM's prompt_token_ids path).
"""

from __future__ import annotations

from typing import List, Sequence, Union

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
OFFSET = 3  # byte b -> id b + OFFSET; ids [3, 258]


def encode(text: str, add_bos: bool = True) -> List[int]:
    ids = [b + OFFSET for b in text.encode("utf-8")]
    return ([BOS_ID] + ids) if add_bos else ids


def decode(ids: Sequence[int]) -> str:
    data = bytes(i - OFFSET for i in ids if OFFSET <= i < OFFSET + 256)
    return data.decode("utf-8", errors="replace")


def to_token_ids(prompt: Union[str, Sequence[int]]) -> List[int]:
    """Accept either text or a ready token-id list (both OpenAI-ish shapes)."""
    if isinstance(prompt, str):
        return encode(prompt)
    return [int(t) for t in prompt]
