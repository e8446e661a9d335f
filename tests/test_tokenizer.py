"""Tokenizer: vendored tokenizer.json (HF fast format) and byte fallback.

The environment has no network, so the test builds a tiny BPE
tokenizer.json programmatically with the offline `tokenizers` wheel —
the same file format a vendored Qwen3 checkpoint dir would carry.
"""

import asyncio
import json

import httpx
import pytest

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.models.registry import get_model_config
from fusioninfer_amd.tokenizer import ByteTokenizer, HFTokenizer, get_tokenizer


@pytest.fixture(scope="module")
def tok_file(tmp_path_factory):
    from tokenizers import Tokenizer, models, pre_tokenizers, decoders

    # byte-level BPE with no merges: every byte is a token — small but
    # structurally identical to a real vendored tokenizer.json
    bl = pre_tokenizers.ByteLevel(add_prefix_space=False)
    vocab = {ch: i for i, ch in enumerate(
        sorted(pre_tokenizers.ByteLevel.alphabet())
    )}
    vocab["ab"] = len(vocab)          # one merge so encoding != raw bytes
    vocab["<|endoftext|>"] = len(vocab)
    tok = Tokenizer(models.BPE(vocab=vocab, merges=[("a", "b")]))
    tok.pre_tokenizer = bl
    tok.decoder = decoders.ByteLevel()
    path = tmp_path_factory.mktemp("tok") / "tokenizer.json"
    tok.save(str(path))
    return str(path)


def test_byte_fallback_roundtrip():
    t = ByteTokenizer(1024)
    assert t.decode(t.encode("hello world")) == "hello world"
    assert t.eos_token_id is None


def test_hf_tokenizer_roundtrip_and_eos(tok_file):
    t = HFTokenizer(tok_file, vocab_size=1024)
    ids = t.encode("hello, world!")
    assert t.decode(ids) == "hello, world!"
    assert all(isinstance(i, int) for i in ids)
    assert t.eos_token_id is not None
    # model-padded ids decode to empty, not an exception
    assert t.decode_one(1000) == ""
    assert t.decode(ids + [1000]) == "hello, world!"


def test_get_tokenizer_resolution(tok_file, tmp_path):
    import shutil
    # explicit --tokenizer file
    t = get_tokenizer(1024, tokenizer_path=tok_file)
    assert isinstance(t, HFTokenizer)
    # checkpoint dir containing tokenizer.json
    d = tmp_path / "ckpt"
    d.mkdir()
    shutil.copy(tok_file, d / "tokenizer.json")
    assert isinstance(get_tokenizer(1024, model_path=str(d)), HFTokenizer)
    # nothing vendored -> byte fallback
    assert isinstance(get_tokenizer(1024, model_path=str(tmp_path)),
                      ByteTokenizer)
    with pytest.raises(FileNotFoundError):
        get_tokenizer(1024, tokenizer_path=str(tmp_path / "nope"))


def test_server_with_hf_tokenizer(tok_file):
    """End-to-end: server encodes string prompts with the vendored
    tokenizer and guided decoding masks over its multi-char vocab."""
    from fusioninfer_amd.server.api_server import build_app
    from fusioninfer_amd.server.serving import ServingEngine

    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=512, max_model_len=256
        ),
    )
    serving = ServingEngine(cfg, device="cpu")
    tok = HFTokenizer(tok_file, vocab_size=cfg.model.vocab_size)
    app = build_app(serving, "tiny-qwen3", tokenizer=tok)

    async def run():
        async with httpx.AsyncClient(
            transport=httpx.ASGITransport(app=app), base_url="http://t"
        ) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "abc", "max_tokens": 4, "temperature": 0.0},
            )
            assert r.status_code == 200
            # prompt went through the vendored BPE: "abc" -> ["ab", "c"]
            # (the byte fallback would count 3)
            assert r.json()["usage"]["prompt_tokens"] == 2

            r = await c.post(
                "/v1/completions",
                json={"prompt": "pick:", "max_tokens": 16,
                      "temperature": 0.0,
                      "guided_choice": ["yes", "no"]},
            )
            assert r.status_code == 200
            assert r.json()["choices"][0]["text"] in {"yes", "no"}

            # non-ASCII choices: byte-level masks make the multi-byte
            # option reachable and the detokenizer reassembles it
            r = await c.post(
                "/v1/completions",
                json={"prompt": "drink:", "max_tokens": 16,
                      "temperature": 0.0,
                      "guided_choice": ["café", "thé"]},
            )
            assert r.status_code == 200
            assert r.json()["choices"][0]["text"] in {"café", "thé"}

    try:
        asyncio.run(run())
    finally:
        serving.shutdown()


def test_chat_template_rendering(tok_file, tmp_path):
    """tokenizer_config.json chat_template renders via jinja2; fallback
    is the flat role-prefixed transcript."""
    import shutil

    d = tmp_path / "ckpt_tmpl"
    d.mkdir()
    shutil.copy(tok_file, d / "tokenizer.json")
    (d / "tokenizer_config.json").write_text(json.dumps({
        "chat_template": (
            "{% for m in messages %}<|{{ m.role }}|>{{ m.content }}"
            "{{ eos_token }}{% endfor %}"
            "{% if add_generation_prompt %}<|assistant|>{% endif %}"
        ),
        "eos_token": "<|endoftext|>",
    }))
    t = get_tokenizer(1024, model_path=str(d))
    out = t.apply_chat_template(
        [{"role": "user", "content": "hi"},
         {"role": "assistant", "content": "yo"}]
    )
    assert out == ("<|user|>hi<|endoftext|><|assistant|>yo<|endoftext|>"
                   "<|assistant|>")
    # eos from tokenizer_config resolves to a real id
    assert t.eos_token_id is not None

    # no template -> fallback
    t2 = ByteTokenizer(1024)
    assert t2.apply_chat_template([{"role": "user", "content": "x"}]) \
        == "user: x"

    # template errors surface as ValueError (mapped to HTTP 400)
    t.chat_template = "{{ raise_exception('nope') }}"
    import pytest as _pytest
    with _pytest.raises(ValueError):
        t.apply_chat_template([])


def test_incremental_detokenizer_multibyte(tok_file):
    """A character whose UTF-8 bytes span several BPE tokens must stream
    out whole, never as U+FFFD fragments (ADVICE round-1 medium)."""
    from fusioninfer_amd.tokenizer import IncrementalDetokenizer

    t = HFTokenizer(tok_file, vocab_size=1024)
    text = "héllo 中文 🙂 ok"
    ids = t.encode(text)
    assert len(ids) > len(text)  # multi-byte chars really did split
    detok = IncrementalDetokenizer(t)
    out = ""
    for i in ids:
        piece = detok.push(i)
        assert "�" not in piece
        out += piece
    out += detok.flush()
    assert out == text
    # per-token independent decode WOULD have mangled it
    assert "�" in "".join(t.decode_one(i) for i in ids)


def test_incremental_detokenizer_byte_fallback():
    from fusioninfer_amd.tokenizer import IncrementalDetokenizer

    t = ByteTokenizer(1024)
    text = "中🙂a"
    ids = t.encode(text)
    detok = IncrementalDetokenizer(t)
    out = ""
    for i in ids:
        piece = detok.push(i)
        assert "�" not in piece
        out += piece
    out += detok.flush()
    assert out == text


def test_incremental_detokenizer_flush_mid_char():
    """Stream ending inside a character: flush emits the honest
    replacement char instead of dropping the tail."""
    from fusioninfer_amd.tokenizer import IncrementalDetokenizer

    t = ByteTokenizer(1024)
    ids = t.encode("中")[:2]  # 2 of 3 bytes
    detok = IncrementalDetokenizer(t)
    assert "".join(detok.push(i) for i in ids) == ""
    assert "�" in detok.flush()


def test_guided_masks_allow_multibyte_utf8(tok_file):
    """Byte-level guided decoding (round-1 advisor, second half): BPE
    tokens carrying PART of a multi-byte UTF-8 char decode to U+FFFD
    through decode_one, which made grammar literals like "café"
    unreachable. With decode_one_bytes + byteized specs the byte
    tokens step the grammar exactly."""
    from fusioninfer_amd.guided import Vocabulary, build_guided

    t = HFTokenizer(tok_file, vocab_size=1024)
    vocab = Vocabulary(1024, t.decode_one_bytes)
    dec = build_guided("choice", ["café", "thé"], vocab)
    ids = t.encode("café")
    assert len(ids) == 5  # c a f + the two é bytes (no-merge tokenizer)
    for tid in ids:
        m = dec.allowed_mask("cpu")
        assert bool(m[tid]), tid
        dec.advance_token(tid)
    assert dec.is_terminal()

    # the OLD char-level vocabulary dead-ends after "caf": the é byte
    # tokens decode to U+FFFD and never match the literal
    vocab_old = Vocabulary(1024, t.decode_one)
    dec_old = build_guided("choice", ["café", "thé"], vocab_old)
    for tid in ids[:3]:
        dec_old.advance_token(tid)
    m = dec_old.allowed_mask("cpu")
    assert m is None or not bool(m[ids[3]])


def test_guided_json_multibyte_string_content(tok_file):
    """json_object mode: CJK bytes are legal JSON-string content at the
    byte level (three continuation bytes step the pushdown grammar)."""
    from fusioninfer_amd.guided import (
        GuidedMaskCache, JsonGrammar, Vocabulary,
    )

    t = HFTokenizer(tok_file, vocab_size=1024)
    vocab = Vocabulary(1024, t.decode_one_bytes)
    g = JsonGrammar(root_object=False)
    cache = GuidedMaskCache(g, vocab)
    st = g.initial()
    for tid in t.encode('"中文"'):
        m = cache.mask(st, "cpu")
        assert bool(m[tid]), tid
        for ch in vocab.strings[tid]:
            st = g.step(st, ch)
    assert g.is_complete(st)
