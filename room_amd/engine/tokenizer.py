"""Self-contained synthetic tokenizer (Qwen3-shaped vocab, 151936 ids).

There is no network to fetch the real Qwen tokenizer files, and the model
weights are random-init anyway (BASELINE: synthetic prompts / random weights).
This tokenizer is deterministic, reversible enough for logging, and produces
realistic token-per-character ratios (whitespace-split words + hash→id) so
prompt token counts match real workloads in shape. Special ids mirror the
Qwen3 chat-template structure (<|im_start|> ... <|im_end|>).
"""
from __future__ import annotations

import hashlib
import re

VOCAB_SIZE = 151936

# special tokens at the top of the vocab (Qwen convention puts them past 151k)
IM_START = 151644
IM_END = 151645
EOS = 151643
TOOL_CALL_START = 151657
TOOL_CALL_END = 151658
_SPECIALS = {IM_START: "<|im_start|>", IM_END: "<|im_end|>", EOS: "<|endoftext|>",
             TOOL_CALL_START: "<tool_call>", TOOL_CALL_END: "</tool_call>"}

_WORD_RE = re.compile(r"\S+|\s+")
_LITERAL_RE = re.compile(r"^⟨(\d+)⟩$")  # decode() output round-trips to the id
_ID_SPACE = 151000  # hash ids land below the specials
_NAME_TO_ID = {v: k for k, v in _SPECIALS.items()}


def _word_id(word: str) -> int:
    h = hashlib.blake2s(word.encode(), digest_size=4).digest()
    return int.from_bytes(h, "little") % _ID_SPACE


def encode(text: str) -> list[int]:
    ids = []
    for m in _WORD_RE.finditer(text):
        tok = m.group(0)
        if tok.isspace():
            continue  # whitespace folds into word boundaries (≈1 token/word)
        lit = _LITERAL_RE.match(tok)
        if lit:  # re-encoding previously generated text reproduces the ids,
            ids.append(int(lit.group(1)) % VOCAB_SIZE)  # keeping KV prefixes valid
            continue
        if tok in _NAME_TO_ID:
            ids.append(_NAME_TO_ID[tok])
            continue
        # long words split into 6-char chunks (mimics BPE token/char ratio)
        for i in range(0, len(tok), 6):
            ids.append(_word_id(tok[i:i + 6]))
    return ids


def decode(ids: list[int]) -> str:
    parts = []
    for i in ids:
        if i in _SPECIALS:
            parts.append(_SPECIALS[i])
        else:
            parts.append(f"⟨{i}⟩")
    return " ".join(parts)


def encode_chat(messages: list[dict]) -> list[int]:
    """Qwen3-shaped chat template: <|im_start|>role\\ncontent<|im_end|> per turn,
    then the assistant generation header."""
    ids: list[int] = []
    for m in messages:
        ids.append(IM_START)
        ids.extend(encode(str(m.get("role", "user"))))
        ids.extend(encode(str(m.get("content", ""))))
        ids.append(IM_END)
    ids.append(IM_START)
    ids.extend(encode("assistant"))
    return ids
