"""Tokenizers for the engine.

- HF tokenizer (tokenizer.json in the model dir) when available.
- SyntheticTokenizer otherwise: deterministic word-hash tokens. Identical
  text prefixes map to identical token prefixes, so prefix caching and
  CHWBL prefix routing behave exactly as with a real tokenizer — this is
  what bench/k6-style runs use (no network for real tokenizer files).
"""
from __future__ import annotations

import os
import zlib


class SyntheticTokenizer:
    """Word-hash tokenizer plus a printable-ASCII char region (ids 3..97
    decode to chr(32..126)) so character-level features — JSON-mode
    constrained decoding, stop strings with punctuation — are exercisable
    without a real tokenizer file."""

    CHAR_LO, CHAR_HI = 3, 97  # id -> chr(id + 29): ' ' .. '~'

    def __init__(self, vocab_size: int, bos_token_id: int = 1, eos_token_id: int = 2):
        self.vocab_size = vocab_size
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        # multimodal: "<image>" in the prompt maps to this id when set
        # (the server wires it from the model config; id 99 for presets)
        self.image_token_id: int | None = None
        # word-hash ids start above the specials + char region
        self._lo = 100

    def char_token(self, ch: str) -> int:
        o = ord(ch)
        assert 32 <= o <= 126, ch
        return o - 29

    def encode(self, text: str, add_bos: bool = False) -> list[int]:
        toks: list[int] = [self.bos_token_id] if add_bos else []
        span = self.vocab_size - self._lo
        for word in text.split():
            if word == "<image>" and self.image_token_id is not None:
                toks.append(self.image_token_id)
                continue
            toks.append(self._lo + zlib.crc32(word.encode()) % span)
        return toks

    def decode(self, token_ids: list[int]) -> str:
        out: list[str] = []
        for t in token_ids:
            if self.CHAR_LO <= t <= self.CHAR_HI:
                out.append(chr(t + 29))
            else:
                if out:
                    out.append(" ")
                out.append(f"t{t}")
        return "".join(out)


class HFTokenizer:
    def __init__(self, path: str):
        from transformers import AutoTokenizer

        self._tok = AutoTokenizer.from_pretrained(path)
        self.vocab_size = self._tok.vocab_size
        self.bos_token_id = self._tok.bos_token_id
        self.eos_token_id = self._tok.eos_token_id

    def encode(self, text: str, add_bos: bool = False) -> list[int]:
        return self._tok.encode(text, add_special_tokens=add_bos)

    def decode(self, token_ids: list[int]) -> str:
        return self._tok.decode(token_ids, skip_special_tokens=True)


class StreamDecoder:
    """Incremental detokenizer: O(window + new tokens) per push instead of
    re-decoding the whole output every token (the r1 stop-string path was
    O(n^2) in generation length). vLLM-style prefix/read offsets: each
    decode call covers only the tokens since the last committed text, with
    one already-committed anchor token so separator-dependent decoders
    (sentencepiece spaces, byte-level merges) produce exact deltas; a
    trailing U+FFFD holds the delta until the multi-byte sequence closes.
    """

    def __init__(self, tok):
        self.tok = tok
        self.ids: list[int] = []
        self.text = ""
        self._prefix = 0  # decode-window start (text before it committed)
        self._read = 0    # ids[:_read] are reflected in self.text

    def push(self, new_ids: list[int]) -> str:
        if new_ids:
            self.ids.extend(new_ids)
            anchor = self.tok.decode(self.ids[self._prefix : self._read])
            full = self.tok.decode(self.ids[self._prefix :])
            if not full.endswith("�"):
                if len(full) > len(anchor):
                    self.text += full[len(anchor) :]
                    self._prefix = self._read
                self._read = len(self.ids)
        return self.text


def load_tokenizer(model: str, vocab_size: int, bos: int, eos: int):
    if os.path.isdir(model) and (
        os.path.exists(os.path.join(model, "tokenizer.json"))
        or os.path.exists(os.path.join(model, "tokenizer.model"))
    ):
        try:
            return HFTokenizer(model)
        except Exception:
            pass
    return SyntheticTokenizer(vocab_size, bos, eos)


def apply_chat_template(tokenizer, messages: list[dict]) -> list[int]:
    """Deterministic minimal chat template (role tag + content per message).

    Reference analog: the engine containers own their chat template
    (engine contract); format is internal.
    """
    def text_of(m: dict) -> str:
        content = m.get("content") or ""
        if isinstance(content, list):  # OpenAI content-parts form
            parts: list[str] = []
            for p in content:
                if not isinstance(p, dict):
                    continue
                if p.get("type") == "image_url":
                    # marker consumed by the tokenizer (vision models) or
                    # left as plain text (text-only models reject images
                    # at admission)
                    parts.append("<image>")
                else:
                    parts.append(p.get("text", ""))
            content = " ".join(parts)
        return content

    # real checkpoints: use the model's own chat template
    tk = getattr(tokenizer, "_tok", None)
    if tk is not None and getattr(tk, "chat_template", None):
        try:
            return tk.apply_chat_template(
                [{"role": m.get("role", "user"), "content": text_of(m)}
                 for m in messages],
                add_generation_prompt=True,
            )
        except Exception:
            pass
    role_ids = {"system": 3, "user": 4, "assistant": 5, "tool": 6}
    toks: list[int] = [tokenizer.bos_token_id]
    for m in messages:
        toks.append(role_ids.get(m.get("role", "user"), 4))
        toks.extend(tokenizer.encode(text_of(m)))
    toks.append(role_ids["assistant"])  # generation prompt
    return toks
