"""Tokenizers for the frontend preprocessor.

ByteTokenizer: deterministic byte-level tokenizer (no network, no files) —
the default for synthetic/random-weight serving and tests.
HFTokenizer: wraps a local `tokenizers` JSON file when one exists.
Chat templating is jinja2 with a Llama-3-style default template
(the reference renders chat templates via minijinja —
preprocessor.rs OpenAIPreprocessor; ours is the Python analog).
"""
from __future__ import annotations

from typing import List, Optional

import jinja2

DEFAULT_CHAT_TEMPLATE = (
    "{% for m in messages %}"
    "<|start_header_id|>{{ m.role }}<|end_header_id|>\n\n{{ m.content }}<|eot_id|>"
    "{% endfor %}"
    "{% if add_generation_prompt %}"
    "<|start_header_id|>assistant<|end_header_id|>\n\n"
    "{% endif %}"
)


class ByteTokenizer:
    """Bytes + special tokens. ids: 0..255 = bytes; 256=bos, 257=eos,
    258=pad; vocab padded to `vocab_size`."""

    def __init__(self, vocab_size: int = 512):
        self.vocab_size = max(vocab_size, 260)
        self.bos_id = 256
        self.eos_id = 257
        self.pad_id = 258

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = list(text.encode("utf-8"))
        return ([self.bos_id] if add_bos else []) + ids

    def decode(self, ids: List[int]) -> str:
        bs = bytes(i for i in ids if 0 <= i < 256)
        try:
            return bs.decode("utf-8")
        except UnicodeDecodeError:
            # non-UTF-8 byte streams (synthetic tokens): latin-1 maps every
            # byte to a DISTINCT char, keeping decode stable and reversible
            # (utf-8 replacement chars collapse distinct bytes and break
            # incremental/full decode consistency)
            return bs.decode("latin-1")

    def decode_incremental(self, ids: List[int], prev_len: int) -> str:
        """Decode the new suffix, robust to split UTF-8 sequences."""
        return self.decode(ids[prev_len:])


class HFTokenizer:
    def __init__(self, path: str):
        from tokenizers import Tokenizer
        self.tk = Tokenizer.from_file(path)
        self.vocab_size = self.tk.get_vocab_size()
        self.eos_id = None
        for cand in ("</s>", "<|eot_id|>", "<|endoftext|>", "<eos>"):
            tid = self.tk.token_to_id(cand)
            if tid is not None:
                self.eos_id = tid
                break
        self.bos_id = self.tk.token_to_id("<s>")

    def encode(self, text: str, add_bos: bool = False) -> List[int]:
        ids = self.tk.encode(text).ids
        return ([self.bos_id] if add_bos and self.bos_id is not None else []) + ids

    def decode(self, ids: List[int]) -> str:
        return self.tk.decode(ids)

    def decode_incremental(self, ids: List[int], prev_len: int) -> str:
        # decode with one token of left context to keep merges stable
        full = self.tk.decode(ids[max(0, prev_len - 1):])
        prevtxt = self.tk.decode(ids[max(0, prev_len - 1):prev_len])
        return full[len(prevtxt):]


def make_tokenizer(spec: Optional[dict]):
    spec = spec or {}
    t = spec.get("type", "byte")
    if t == "byte":
        return ByteTokenizer(spec.get("vocab_size", 512))
    if t == "hf":
        return HFTokenizer(spec["path"])
    raise ValueError(f"unknown tokenizer type {t!r}")


class ChatTemplater:
    def __init__(self, template: Optional[str] = None):
        self.env = jinja2.Environment()
        self.template = self.env.from_string(template or DEFAULT_CHAT_TEMPLATE)

    def render(self, messages: List[dict],
               add_generation_prompt: bool = True) -> str:
        return self.template.render(messages=messages,
                                    add_generation_prompt=add_generation_prompt)
