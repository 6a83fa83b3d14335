"""Front-door tokenization: byte-level codec + optional real-vocabulary backend.

The reference proxies prompt text straight to vLLM, which owns tokenization.
This framework serves random-init models on synthetic workloads (no network,
no checkpoint vocabularies), so the DEFAULT is a deterministic byte-level
codec: UTF-8 byte ``b`` maps to token id ``b + OFFSET``.  It round-trips any
string, needs no assets, and keeps ids well inside every model's vocab.
Requests may also bypass it by passing ``prompt`` as a list of token ids
(mirrors vLLM's prompt_token_ids path).

A REAL vocabulary (the vLLM API surface the reference pins) plugs in via
``load_backend(path)`` — any HuggingFace ``tokenizer.json`` (the
``tokenizers`` library ships in the image; only the asset file is absent
offline) — or ``set_backend(tok)`` for an in-process tokenizer object.
``python -m llm_instance_gateway_amd.server --tokenizer path/tokenizer.json``
wires it for the whole front door.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Union

PAD_ID = 0
BOS_ID = 1
EOS_ID = 2
OFFSET = 3  # byte b -> id b + OFFSET; ids [3, 258]

_BACKEND = None  # tokenizers.Tokenizer (or API-compatible), None = byte codec


def set_backend(tok) -> None:
    """Install a real tokenizer (must expose .encode(str).ids and
    .decode(list[int]))."""
    global _BACKEND
    _BACKEND = tok


def load_backend(path: str) -> bool:
    """Load a HuggingFace tokenizer.json; returns False if unavailable."""
    try:
        from tokenizers import Tokenizer

        set_backend(Tokenizer.from_file(path))
        return True
    except Exception:
        return False


def backend_name() -> str:
    return type(_BACKEND).__name__ if _BACKEND is not None else "byte-codec"


def has_backend() -> bool:
    """True when a real (subword) tokenizer is installed.  Under the byte
    codec token-id stop matching is exactly string matching (1 byte = 1 id,
    context-free), so decoded-text stop checks are redundant."""
    return _BACKEND is not None


def encode(text: str, add_bos: bool = True) -> List[int]:
    if _BACKEND is not None:
        return list(_BACKEND.encode(text).ids)
    ids = [b + OFFSET for b in text.encode("utf-8")]
    return ([BOS_ID] + ids) if add_bos else ids


def decode(ids: Sequence[int]) -> str:
    if _BACKEND is not None:
        return _BACKEND.decode(list(ids))
    data = bytes(i - OFFSET for i in ids if OFFSET <= i < OFFSET + 256)
    return data.decode("utf-8", errors="replace")


def apply_chat_template(messages: Sequence, add_generation_prompt: bool = True) -> str:
    """Minimal chat template (Llama-3-instruct-shaped header tokens as text;
    random-init weights make the exact special-token ids immaterial, but the
    structure matches what /v1/chat/completions clients expect)."""
    parts = []
    for m in messages:
        role = m.role if hasattr(m, "role") else m["role"]
        content = m.content if hasattr(m, "content") else m["content"]
        parts.append(f"<|start_header_id|>{role}<|end_header_id|>\n\n{content}<|eot_id|>")
    if add_generation_prompt:
        parts.append("<|start_header_id|>assistant<|end_header_id|>\n\n")
    return "".join(parts)


def to_token_ids(prompt: Union[str, Sequence[int]]) -> List[int]:
    """Accept either text or a ready token-id list (both OpenAI-ish shapes)."""
    if isinstance(prompt, str):
        return encode(prompt)
    return [int(t) for t in prompt]
