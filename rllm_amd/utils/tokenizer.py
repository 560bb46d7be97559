"""Tokenizer interface + a hermetic byte-level tokenizer.

The engine accepts any HuggingFace-compatible tokenizer (encode/decode/
apply_chat_template). There is no network in this environment, so the
in-repo ByteTokenizer (256 byte tokens + special tokens) makes the whole
chat → tokens → chat path runnable hermetically in tests, CPU demos, and
the gateway integration suite; real runs pass a local HF tokenizer dir.
"""

from __future__ import annotations


class ByteTokenizer:
    """Byte-level tokenizer: token i == byte i for i < 256; specials above."""

    SPECIALS = ["<|pad|>", "<|im_start|>", "<|im_end|>", "<|endoftext|>"]

    def __init__(self):
        self.vocab_size = 256 + len(self.SPECIALS)
        self._special_to_id = {s: 256 + i for i, s in enumerate(self.SPECIALS)}
        self._id_to_special = {v: k for k, v in self._special_to_id.items()}
        self.pad_token_id = self._special_to_id["<|pad|>"]
        self.im_start_id = self._special_to_id["<|im_start|>"]
        self.im_end_id = self._special_to_id["<|im_end|>"]
        self.eos_token_id = self._special_to_id["<|im_end|>"]
        self.eos_token = "<|im_end|>"

    def encode(self, text: str, add_special_tokens: bool = False) -> list[int]:
        out: list[int] = []
        i = 0
        while i < len(text):
            matched = False
            for s, tid in self._special_to_id.items():
                if text.startswith(s, i):
                    out.append(tid)
                    i += len(s)
                    matched = True
                    break
            if not matched:
                out.extend(text[i].encode("utf-8"))
                i += 1
        return out

    def decode(self, ids: list[int], skip_special_tokens: bool = False) -> str:
        parts: list[str] = []
        byte_buf = bytearray()
        for t in ids:
            if t < 256:
                byte_buf.append(t)
            else:
                if byte_buf:
                    parts.append(byte_buf.decode("utf-8", errors="replace"))
                    byte_buf = bytearray()
                if not skip_special_tokens:
                    parts.append(self._id_to_special.get(t, ""))
        if byte_buf:
            parts.append(byte_buf.decode("utf-8", errors="replace"))
        return "".join(parts)

    def __call__(self, text: str, **kw):
        return {"input_ids": self.encode(text)}


def load_tokenizer(name_or_path: str | None):
    """Local HF tokenizer dir, or the hermetic byte tokenizer."""
    if name_or_path in (None, "byte", "bytes"):
        return ByteTokenizer()
    from transformers import AutoTokenizer

    return AutoTokenizer.from_pretrained(name_or_path, trust_remote_code=True)
