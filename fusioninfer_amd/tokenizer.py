"""Tokenization for the OpenAI server.

The environment has no network access, so tokenizer files can only come
vendored with a checkpoint directory (`tokenizer.json`, the HF fast-
tokenizer format, loaded with the offline `tokenizers` wheel). Without
one, string prompts fall back to a reversible byte-level scheme — ids
offset by 3 so they stay in-vocab — which the benchmark and EPP flows
use (they submit token-id prompts anyway).

Capability parity: the reference's engine containers mount HF tokenizer
files next to the weights (SURVEY.md §2.3 `--model Qwen/Qwen3-8B`); this
module gives the same behavior for local checkpoint dirs.
"""

from __future__ import annotations

import os
from typing import List, Optional, Union


def _render_chat_template(template: str, messages, bos: str, eos: str,
                          add_generation_prompt: bool) -> str:
    """HF-style chat template rendering (jinja2, sandboxed env with the
    helpers templates expect: raise_exception, tojson)."""
    from jinja2.sandbox import ImmutableSandboxedEnvironment

    env = ImmutableSandboxedEnvironment(trim_blocks=True, lstrip_blocks=True)

    def raise_exception(msg):
        raise ValueError(f"chat template error: {msg}")

    env.globals["raise_exception"] = raise_exception
    return env.from_string(template).render(
        messages=messages,
        bos_token=bos,
        eos_token=eos,
        add_generation_prompt=add_generation_prompt,
    )


def default_chat_format(messages) -> str:
    """Template-less fallback: the flat role-prefixed transcript."""
    text = "\n".join(
        f"{m.get('role', 'user')}: {m.get('content', '')}" for m in messages
    )
    return text


_BYTE_DECODER: Optional[dict] = None


def _byte_level_decoder() -> dict:
    """char -> byte map of the byte-level BPE alphabet (the GPT-2 table
    used by Qwen/Llama tokenizer.json ByteLevel pre-tokenizers)."""
    global _BYTE_DECODER
    if _BYTE_DECODER is None:
        bs = (list(range(33, 127)) + list(range(161, 173))
              + list(range(174, 256)))
        cs = bs[:]
        n = 0
        for b in range(256):
            if b not in bs:
                bs.append(b)
                cs.append(256 + n)
                n += 1
        _BYTE_DECODER = {chr(c): b for b, c in zip(bs, cs)}
    return _BYTE_DECODER


class ByteTokenizer:
    """Reversible byte-level fallback (no vocab files needed)."""

    def __init__(self, vocab_size: int):
        self.vocab_size = vocab_size
        self.eos_token_id: Optional[int] = None
        self.chat_template: Optional[str] = None

    def apply_chat_template(self, messages,
                            add_generation_prompt: bool = True) -> str:
        if self.chat_template:
            return _render_chat_template(
                self.chat_template, messages, "", "", add_generation_prompt
            )
        return default_chat_format(messages)

    def encode(self, text: str) -> List[int]:
        return [min(b + 3, self.vocab_size - 1) for b in text.encode("utf-8")]

    def decode(self, token_ids: List[int]) -> str:
        return bytes(max(t - 3, 0) & 0xFF for t in token_ids).decode(
            "utf-8", errors="replace"
        )

    def decode_one(self, token_id: int) -> str:
        return bytes([max(token_id - 3, 0) & 0xFF]).decode(
            "utf-8", errors="replace"
        )

    def decode_one_bytes(self, token_id: int) -> str:
        """Per-token RAW byte as a latin-1 char (byte<->char bijection).
        Guided-decoding masks step grammars at byte level with this, so
        bytes >= 0x80 stay representable (decode_one turns them into
        U+FFFD, which silently masked non-ASCII out of grammars —
        round-1 advisor finding, second half)."""
        return chr(max(token_id - 3, 0) & 0xFF)


class HFTokenizer:
    """tokenizer.json (HF fast format) via the `tokenizers` library."""

    def __init__(self, path: str, vocab_size: int):
        from tokenizers import Tokenizer

        self._tok = Tokenizer.from_file(path)
        self.vocab_size = vocab_size
        # model vocab may be padded past the tokenizer's (Qwen3 pads
        # 151669 -> 151936); ids past the tokenizer's range decode to ""
        self._n = self._tok.get_vocab_size(with_added_tokens=True)
        self.eos_token_id = None
        self.chat_template: Optional[str] = None
        self._eos_token = ""
        self._bos_token = ""
        for name in ("</s>", "<|endoftext|>", "<|im_end|>", "<eos>"):
            tid = self._tok.token_to_id(name)
            if tid is not None:
                self.eos_token_id = tid
                self._eos_token = name
                break
        # tokenizer_config.json next to tokenizer.json carries the chat
        # template + canonical special tokens (HF layout)
        cfg_path = os.path.join(os.path.dirname(path),
                                "tokenizer_config.json")
        if os.path.isfile(cfg_path):
            import json as _json

            try:
                with open(cfg_path) as f:
                    tc = _json.load(f)
            except ValueError:
                tc = {}
            tmpl = tc.get("chat_template")
            if isinstance(tmpl, str):
                self.chat_template = tmpl
            for key, attr in (("eos_token", "_eos_token"),
                              ("bos_token", "_bos_token")):
                v = tc.get(key)
                if isinstance(v, dict):
                    v = v.get("content")
                if isinstance(v, str):
                    setattr(self, attr, v)
                    if key == "eos_token":
                        tid = self._tok.token_to_id(v)
                        if tid is not None:
                            self.eos_token_id = tid

    def apply_chat_template(self, messages,
                            add_generation_prompt: bool = True) -> str:
        if self.chat_template:
            return _render_chat_template(
                self.chat_template, messages, self._bos_token,
                self._eos_token, add_generation_prompt,
            )
        return default_chat_format(messages)

    def encode(self, text: str) -> List[int]:
        return self._tok.encode(text, add_special_tokens=False).ids

    def decode(self, token_ids: List[int]) -> str:
        ids = [t for t in token_ids if 0 <= t < self._n]
        return self._tok.decode(ids, skip_special_tokens=False)

    def decode_one(self, token_id: int) -> str:
        # per-token text for streaming deltas and guided-decoding masks
        if not 0 <= token_id < self._n:
            return ""
        return self._tok.decode([token_id], skip_special_tokens=False)

    def decode_one_bytes(self, token_id: int) -> str:
        """Per-token RAW BYTES as a latin-1 string: a byte-level BPE
        token that carries part of a multi-byte UTF-8 character decodes
        to U+FFFD through decode_one, which excluded every such token
        from guided-decoding masks (no CJK/emoji under guided JSON).
        Mapping token text back through the byte-level alphabet keeps
        the exact bytes; grammars then run at byte level (the engine's
        detokenizer reassembles characters on output)."""
        if not 0 <= token_id < self._n:
            return ""
        tok = self._tok.id_to_token(token_id)
        if tok is None:
            return ""
        inv = _byte_level_decoder()
        try:
            return bytes(inv[c] for c in tok).decode("latin-1")
        except KeyError:
            # non-byte-level token (e.g. an added special token): its
            # literal text is the right grammar-visible form
            return self.decode_one(token_id)


class IncrementalDetokenizer:
    """Streaming detokenization that never splits a multi-byte UTF-8
    character across deltas.

    decode_one() decodes each token independently, so a character whose
    bytes span two BPE tokens (CJK/emoji under byte-level BPE) comes out
    as U+FFFD. This class instead decodes a sliding window of recent ids
    and emits only the stable suffix: while the window's decode ends in
    U+FFFD (an incomplete byte sequence) the text is held back until the
    completing token arrives. Mirrors the reference engines' incremental
    detokenization contract (the vLLM images the reference launches,
    SURVEY.md §2.3).
    """

    #: tokens of context kept before the unread window (byte-level BPE
    #: decoders are concatenative, but sentencepiece-style decoders join
    #: with context; a few tokens of prefix keeps deltas exact)
    _CTX = 6

    def __init__(self, tok, initial_ids=()):
        self.tok = tok
        self.ids: List[int] = list(initial_ids)
        self.prefix_offset = max(len(self.ids) - self._CTX, 0)
        self.read_offset = len(self.ids)

    def push(self, token_id: int) -> str:
        """Add one generated token; return the newly-stable text ("" if
        the tail is still an incomplete character)."""
        self.ids.append(int(token_id))
        prefix = self.tok.decode(self.ids[self.prefix_offset:self.read_offset])
        full = self.tok.decode(self.ids[self.prefix_offset:])
        if len(full) > len(prefix) and not full.endswith("�"):
            delta = full[len(prefix):]
            self.prefix_offset = max(len(self.ids) - self._CTX, 0)
            self.read_offset = len(self.ids)
            return delta
        return ""

    def flush(self) -> str:
        """Emit any held-back tail (stream ended mid-character: the
        replacement char is then the honest output)."""
        prefix = self.tok.decode(self.ids[self.prefix_offset:self.read_offset])
        full = self.tok.decode(self.ids[self.prefix_offset:])
        self.read_offset = len(self.ids)
        self.prefix_offset = max(len(self.ids) - self._CTX, 0)
        return full[len(prefix):]


Tok = Union[ByteTokenizer, HFTokenizer]


def get_tokenizer(
    vocab_size: int,
    model_path: Optional[str] = None,
    tokenizer_path: Optional[str] = None,
) -> Tok:
    """tokenizer.json from --tokenizer, else from the checkpoint dir,
    else the byte fallback."""
    candidates = []
    if tokenizer_path:
        candidates.append(tokenizer_path)
        candidates.append(os.path.join(tokenizer_path, "tokenizer.json"))
    if model_path:
        candidates.append(os.path.join(model_path, "tokenizer.json"))
    for c in candidates:
        if os.path.isfile(c):
            return HFTokenizer(c, vocab_size)
    if tokenizer_path:
        raise FileNotFoundError(
            f"--tokenizer {tokenizer_path!r}: no tokenizer.json found"
        )
    return ByteTokenizer(vocab_size)
