"""Guided decoding: regex-derivative engine, JSON pushdown grammar,
schema->regex compilation, and engine-level grammar-masked sampling.

Parity target: vLLM structured outputs (response_format json_object,
guided_json / guided_regex / guided_choice).
"""

import json

import pytest
import torch

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams
from fusioninfer_amd.guided import (
    GuidedMaskCache,
    JsonGrammar,
    RegexError,
    RegexGrammar,
    Vocabulary,
    build_guided,
    schema_to_regex,
)
from fusioninfer_amd.models.registry import get_model_config


def accepts(g, s: str) -> bool:
    st = g.initial()
    for ch in s:
        st = g.step(st, ch)
        if st is None:
            return False
    return g.is_complete(st)


def viable(g, s: str) -> bool:
    st = g.initial()
    for ch in s:
        st = g.step(st, ch)
        if st is None:
            return False
    return True


# ------------------------------------------------------------ regex engine
def test_regex_literals_and_classes():
    g = RegexGrammar("ab[0-9]+z?")
    assert accepts(g, "ab3")
    assert accepts(g, "ab123z")
    assert not accepts(g, "ab")
    assert not accepts(g, "abz")
    assert viable(g, "ab1") and not viable(g, "b")


def test_regex_alt_group_repeat():
    g = RegexGrammar("(yes|no|maybe)")
    assert accepts(g, "yes") and accepts(g, "no") and accepts(g, "maybe")
    assert not accepts(g, "ye") and viable(g, "ye")
    g2 = RegexGrammar("a{2,4}")
    assert not accepts(g2, "a") and accepts(g2, "aa") and accepts(g2, "aaaa")
    assert not viable(g2, "aaaaa")
    g3 = RegexGrammar("(?:ab)+")
    assert accepts(g3, "abab") and not accepts(g3, "aba")


def test_regex_escapes_dot_negation():
    g = RegexGrammar(r"\d\d-\w+\s?.")
    assert accepts(g, "12-ok x")
    assert accepts(g, "12-ok!")
    assert not accepts(g, "1a-ok x")
    g2 = RegexGrammar(r"[^abc]x")
    assert accepts(g2, "dx") and not accepts(g2, "ax")


def test_regex_errors():
    with pytest.raises(RegexError):
        RegexGrammar("(unclosed")
    with pytest.raises(RegexError):
        RegexGrammar("*dangling")
    with pytest.raises(RegexError):
        RegexGrammar("a{999999}")


# ------------------------------------------------------------ JSON grammar
def test_json_accepts_wellformed():
    g = JsonGrammar(root_object=True)
    for s in [
        '{}',
        '{"a":1}',
        '{"a":-1.5e3,"b":[true,false,null]}',
        '{"s":"he\\"llo","n":{"x":[]}}',
        '{"u":"\\u00e9"}',
        '{ "a" : [ 1 , 2 ] }',
    ]:
        assert accepts(g, s), s


def test_json_rejects_malformed():
    g = JsonGrammar(root_object=True)
    for s in [
        '[1,2]',            # root must be an object in json_object mode
        '{"a":01}',         # leading zero
        '{"a":1,}',         # trailing comma
        '{"a" 1}',          # missing colon
        '{"a":tru}',
        "{'a':1}",
    ]:
        assert not accepts(g, s), s
    # dead prefixes
    assert not viable(g, '{"a":,')
    assert not viable(g, 'x')


def test_json_any_root_and_termination():
    g = JsonGrammar(root_object=False)
    assert accepts(g, '[1,2,3]') and accepts(g, '"str"') and accepts(g, 'true')
    # after a complete root container nothing may follow (termination)
    st = g.initial()
    for ch in '{"a":1}':
        st = g.step(st, ch)
    assert g.is_complete(st) and not g.can_extend(st)
    assert g.step(st, " ") is None


def test_json_signature_depth_independent():
    """Mask-cache signatures collapse by stack top, not depth."""
    g = JsonGrammar(root_object=False)
    s1, s2 = g.initial(), g.initial()
    for ch in '{"a":':
        s1 = g.step(s1, ch)
    for ch in '{"a":{"b":{"c":':
        s2 = g.step(s2, ch)
    assert g.signature(s1) == g.signature(s2)


# --------------------------------------------------------- schema -> regex
def test_schema_to_regex_object():
    pattern = schema_to_regex({
        "type": "object",
        "properties": {
            "name": {"type": "string"},
            "age": {"type": "integer"},
            "tags": {"type": "array", "items": {"type": "string"}},
            "ok": {"type": "boolean"},
        },
    })
    g = RegexGrammar(pattern)
    assert accepts(g, '{"name":"bo","age":41,"tags":["x","y"],"ok":true}')
    assert accepts(g, '{"name":"","age":0,"tags":[],"ok":false}')
    assert not accepts(g, '{"age":41}')      # fixed property order/presence
    assert not accepts(g, '{"name":"bo","age":4.5,"tags":[],"ok":true}')


def test_schema_object_optional_properties():
    """Properties outside `required` may be skipped (order stays fixed,
    commas only between present properties)."""
    pattern = schema_to_regex({
        "type": "object",
        "properties": {
            "pre": {"type": "boolean"},       # optional, before required
            "name": {"type": "string"},       # required
            "age": {"type": "integer"},       # optional
            "ok": {"type": "boolean"},        # required
            "tag": {"type": "string"},        # optional, trailing
        },
        "required": ["name", "ok"],
    })
    g = RegexGrammar(pattern)
    assert accepts(g, '{"pre":true,"name":"bo","age":4,"ok":true,"tag":"x"}')
    assert accepts(g, '{"name":"bo","ok":false}')          # optionals skipped
    assert accepts(g, '{"name":"bo","age":7,"ok":true}')
    assert accepts(g, '{"pre":false,"name":"bo","ok":true,"tag":"y"}')
    assert not accepts(g, '{"age":7,"ok":true}')           # missing required
    assert not accepts(g, '{"name":"bo"}')                 # missing required
    assert not accepts(g, '{"age":7,"name":"bo","ok":true}')  # order fixed
    assert not accepts(g, '{"name":"bo","ok":true,}')      # dangling comma


def test_schema_object_all_optional():
    g = RegexGrammar(schema_to_regex({
        "type": "object",
        "properties": {"a": {"type": "integer"}, "b": {"type": "boolean"}},
        "required": [],
    }))
    assert accepts(g, "{}")
    assert accepts(g, '{"a":1}')
    assert accepts(g, '{"b":true}')
    assert accepts(g, '{"a":1,"b":false}')
    assert not accepts(g, '{"b":true,"a":1}')  # order fixed
    assert not accepts(g, '{,}')


def test_schema_enum_and_unsupported():
    g = RegexGrammar(schema_to_regex({"enum": ["red", "green", 3]}))
    assert accepts(g, '"red"') and accepts(g, "3") and not accepts(g, '"blue"')
    with pytest.raises(ValueError):
        schema_to_regex({"type": "object"})  # no properties


# ------------------------------------------------------ vocab + mask cache
def byte_vocab(vocab_size=300):
    # mirrors the server's byte-fallback tokenizer (api_server.decode_tokens)
    return Vocabulary(
        vocab_size,
        lambda t: bytes([max(t - 3, 0) & 0xFF]).decode("utf-8", "replace"),
    )


def test_mask_cache_masks_and_reuse():
    vocab = byte_vocab()
    cache = GuidedMaskCache(RegexGrammar("(yes|no)"), vocab)
    g = cache.grammar
    st = g.initial()
    m = cache.mask(st, "cpu")
    allowed = {vocab.strings[i] for i in torch.nonzero(m).flatten().tolist()}
    assert allowed == {"y", "n"}
    # same signature -> same cached tensor object
    assert cache.mask(g.initial(), "cpu") is m


# ----------------------------------------------------------- engine-level
def make_engine(seed=0):
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=1024, max_model_len=256
        ),
        seed=seed,
    )
    return LLMEngine(cfg, device="cpu")


def decode_bytes(toks):
    return bytes(max(t - 3, 0) & 0xFF for t in toks).decode("utf-8", "replace")


def test_engine_guided_choice():
    torch.manual_seed(0)
    eng = make_engine()
    vocab = byte_vocab(eng.cfg.model.vocab_size)
    for temp in (0.0, 0.9):
        guided = build_guided("choice", ["yes", "no", "maybe"], vocab)
        outs = eng.generate(
            [[5, 6, 7] * 5],
            SamplingParams(max_tokens=20, temperature=temp, guided=guided),
        )
        text = decode_bytes(outs[0].output_token_ids)
        assert text in {"yes", "no", "maybe"}


def test_engine_guided_schema_json():
    torch.manual_seed(0)
    eng = make_engine()
    vocab = byte_vocab(eng.cfg.model.vocab_size)
    guided = build_guided(
        "json_schema",
        {"type": "object", "properties": {"ok": {"type": "boolean"},
                                          "n": {"type": "integer"}}},
        vocab,
    )
    outs = eng.generate(
        [[9, 2, 4] * 6],
        SamplingParams(max_tokens=64, temperature=0.0, guided=guided),
    )
    obj = json.loads(decode_bytes(outs[0].output_token_ids))
    assert set(obj) == {"ok", "n"}
    assert isinstance(obj["ok"], bool) and isinstance(obj["n"], int)


def test_engine_guided_json_object_prefix_always_viable():
    """Generic JSON mode: termination may exceed max_tokens with a
    random-init model, but every emitted prefix must stay inside the
    grammar (mask correctness)."""
    torch.manual_seed(0)
    eng = make_engine()
    vocab = byte_vocab(eng.cfg.model.vocab_size)
    guided = build_guided("json_object", None, vocab)
    outs = eng.generate(
        [[3, 1, 4] * 6],
        SamplingParams(max_tokens=24, temperature=0.7, guided=guided),
    )
    text = decode_bytes(outs[0].output_token_ids)
    assert viable(JsonGrammar(root_object=True), text), text
    assert text.startswith("{") or text.lstrip(" \t\n\r").startswith("{")


def test_engine_guided_regex_bounded():
    torch.manual_seed(0)
    eng = make_engine()
    vocab = byte_vocab(eng.cfg.model.vocab_size)
    guided = build_guided("regex", r"[ab]{3}-\d{2}", vocab)
    outs = eng.generate(
        [[7, 7, 1] * 5],
        SamplingParams(max_tokens=30, temperature=0.0, guided=guided),
    )
    import re as _re

    text = decode_bytes(outs[0].output_token_ids)
    assert _re.fullmatch(r"[ab]{3}-\d{2}", text), text
    # grammar termination ended the request well before max_tokens
    assert len(outs[0].output_token_ids) == 6


def test_schema_extensions():
    """anyOf / const / type lists / string pattern / min-maxItems /
    optional properties — the wider vLLM guided_json surface."""
    g = RegexGrammar(schema_to_regex({
        "anyOf": [{"type": "integer"}, {"type": "boolean"}],
    }))
    assert accepts(g, "42") and accepts(g, "true") and not accepts(g, '"x"')

    g = RegexGrammar(schema_to_regex({"type": ["string", "null"]}))
    assert accepts(g, '"hi"') and accepts(g, "null")

    g = RegexGrammar(schema_to_regex({"const": "red"}))
    assert accepts(g, '"red"') and not accepts(g, '"blue"')

    g = RegexGrammar(schema_to_regex({
        "type": "string", "pattern": "^[a-f]{2}-[0-9]+$",
    }))
    assert accepts(g, '"ab-42"') and not accepts(g, '"zz-1"')

    g = RegexGrammar(schema_to_regex({
        "type": "array", "items": {"type": "integer"},
        "minItems": 2, "maxItems": 3,
    }))
    assert not accepts(g, "[1]")
    assert accepts(g, "[1,2]") and accepts(g, "[1,2,3]")
    assert not accepts(g, "[1,2,3,4]")

    g = RegexGrammar(schema_to_regex({
        "type": "object",
        "properties": {"a": {"type": "integer"}, "b": {"type": "boolean"},
                       "c": {"type": "string"}},
        "required": ["a", "c"],
    }))
    assert accepts(g, '{"a":1,"c":"x"}')
    assert accepts(g, '{"a":1,"b":true,"c":"x"}')  # optional b allowed
    assert not accepts(g, '{"b":true,"c":"x"}')    # required a missing


def test_schema_bounds_refs_and_allof():
    """xgrammar-class schema coverage: $defs/$ref, string min/maxLength,
    integer minimum/maximum, allOf(single), and the fallbacks."""
    import pytest as _p

    from fusioninfer_amd.guided import RegexGrammar, schema_to_regex

    def accepts(rx, s):
        g = RegexGrammar(rx)
        st = g.initial()
        for ch in s:
            st = g.step(st, ch)
            if st is None:
                return False
        return g.is_complete(st)

    # $defs / $ref
    rx = schema_to_regex({
        "$defs": {"port": {"type": "integer", "minimum": 1, "maximum": 9}},
        "type": "object",
        "properties": {"p": {"$ref": "#/$defs/port"}},
    })
    assert accepts(rx, '{"p":7}')
    assert not accepts(rx, '{"p":0}')

    # integer bounds: enumerated range and sign-only
    rx = schema_to_regex({"type": "integer", "minimum": 250, "maximum": 255})
    assert accepts(rx, "255") and not accepts(rx, "249")
    rx = schema_to_regex({"type": "integer", "minimum": 0})
    assert accepts(rx, "0") and accepts(rx, "123") and not accepts(rx, "-3")
    rx = schema_to_regex({"type": "integer", "maximum": -1})
    assert accepts(rx, "-7") and not accepts(rx, "2")
    with _p.raises(ValueError):
        schema_to_regex({"type": "integer", "minimum": -5, "maximum": 10**9})

    # string length bounds
    rx = schema_to_regex({"type": "string", "minLength": 2, "maxLength": 3})
    assert accepts(rx, '"ab"') and accepts(rx, '"abc"')
    assert not accepts(rx, '"a"') and not accepts(rx, '"abcd"')

    # allOf single-branch merge
    rx = schema_to_regex({
        "allOf": [{"type": "integer"}], "minimum": 1, "maximum": 3,
    })
    assert accepts(rx, "2") and not accepts(rx, "4")
    with _p.raises(ValueError):
        schema_to_regex({"allOf": [{"type": "integer"}, {"minimum": 1}]})

    # unresolved / non-local refs fall back
    with _p.raises(ValueError):
        schema_to_regex({"$ref": "#/$defs/missing"})
    with _p.raises(ValueError):
        schema_to_regex({"$ref": "http://x/schema.json"})


# ------------------------------------------------------------ GBNF grammar
def test_gbnf_basics():
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar('root ::= "yes" | "no" | "maybe"')
    assert accepts(g, "yes") and accepts(g, "no") and accepts(g, "maybe")
    assert not accepts(g, "ye") and viable(g, "ye")
    assert not accepts(g, "yesno")
    assert not viable(g, "z")


def test_gbnf_repetition_classes_groups():
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar('root ::= [a-z]+ ("-" [0-9]{2,3})?')
    assert accepts(g, "abc")
    assert accepts(g, "abc-42") and accepts(g, "abc-123")
    assert not accepts(g, "abc-1")        # {2,3} lower bound
    assert not accepts(g, "abc-1234")     # upper bound
    assert not accepts(g, "ABC")
    g2 = GbnfGrammar('root ::= ("ab")* "!"')
    assert accepts(g2, "!") and accepts(g2, "abab!")
    assert not accepts(g2, "aba!")
    g3 = GbnfGrammar('root ::= [^x]+')
    assert accepts(g3, "abc") and not viable(g3, "x")


def test_gbnf_recursive_rules():
    """Nested/recursive rules — the llama.cpp arithmetic-expression
    shape (right recursion + nesting through parens)."""
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar('''
# expression grammar
root  ::= expr
expr  ::= term (("+" | "-") term)*
term  ::= num | "(" expr ")"
num   ::= [0-9]+
''')
    assert accepts(g, "1")
    assert accepts(g, "1+2-3")
    assert accepts(g, "(1+2)-(3+(4-5))")
    assert not accepts(g, "1+")
    assert viable(g, "((")
    assert not viable(g, ")")


def test_gbnf_escapes_and_errors():
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar(r'root ::= "a\nb" | "\x41"')
    assert accepts(g, "a\nb") and accepts(g, "A")
    with pytest.raises(ValueError, match="no 'root'"):
        GbnfGrammar('start ::= "x"')
    with pytest.raises(ValueError, match="undefined rule"):
        GbnfGrammar('root ::= missing')
    with pytest.raises(ValueError, match="left recursion"):
        GbnfGrammar('root ::= root "x" | "y"')


def test_gbnf_engine_masked_generation():
    """End-to-end: guided_grammar constrains engine output (byte
    vocab)."""
    from fusioninfer_amd.guided import GbnfGrammar, GuidedMaskCache

    vocab = byte_vocab()
    cache = GuidedMaskCache(
        GbnfGrammar('root ::= ("ab" | "cd")+'), vocab
    )
    g = cache.grammar
    st = g.initial()
    m = cache.mask(st, "cpu")
    allowed = {vocab.strings[i] for i in torch.nonzero(m).flatten().tolist()}
    assert allowed == {"a", "c"}
    st = g.step(st, "a")
    m = cache.mask(st, "cpu")
    allowed = {vocab.strings[i] for i in torch.nonzero(m).flatten().tolist()}
    assert allowed == {"b"}


def test_gbnf_edge_cases():
    """Parser edges: multiline rules, nested groups with alternates,
    escaped chars in classes, comments, {n} exact repetition."""
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar('''
root ::= kv ("," kv)*        # object-ish list
kv   ::= key "=" val
key  ::= [a-zA-Z_] [a-zA-Z0-9_]*
val  ::= num | "'" [^']* "'"
num  ::= "-"? [0-9]{1,3}
''')
    assert accepts(g, "a=1")
    assert accepts(g, "key_1=-42,b='x y',c=999")
    assert not accepts(g, "1a=2")          # key can't start with digit
    assert not accepts(g, "a=1234")        # {1,3}
    assert not accepts(g, "a=1,")          # trailing comma
    g2 = GbnfGrammar(r'root ::= ("a" | ("b" "c")+) [\-\]x]{2}')
    assert accepts(g2, "a-]")
    assert accepts(g2, "bcbcxx")
    assert not accepts(g2, "a-")           # exactly 2 from the class
    g3 = GbnfGrammar('root ::= "a"{3}')
    assert accepts(g3, "aaa")
    assert not accepts(g3, "aa") and not accepts(g3, "aaaa")


def test_gbnf_long_repetition_bounded_state():
    """Tail-call elision: unbounded repetition must not grow the parse
    stacks (X* desugars right-recursively; without the elision a ~256
    char input tripped the expansion-depth guard)."""
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar('root ::= [a-z]* "!"')
    st = g.initial()
    for _ in range(2000):
        st = g.step(st, "a")
        assert st is not None
        assert max(len(s) for s in st) <= 4  # stacks stay shallow
    st = g.step(st, "!")
    assert g.is_complete(st)
    g2 = GbnfGrammar('root ::= ("ab")+')
    st = g2.initial()
    for _ in range(500):
        st = g2.step(st, "a")
        st = g2.step(st, "b")
    assert g2.is_complete(st)


def test_gbnf_mini_json_grammar():
    """A practical GBNF grammar (mini-JSON: nested arrays/objects of
    ints and short strings) — the shape users actually ship."""
    from fusioninfer_amd.guided import GbnfGrammar

    g = GbnfGrammar(r'''
root   ::= value
value  ::= object | array | number | string
object ::= "{" (pair ("," pair)*)? "}"
pair   ::= string ":" value
array  ::= "[" (value ("," value)*)? "]"
number ::= "-"? [0-9]+
string ::= "\"" [a-z0-9_ ]* "\""
''')
    for ok in ('{}', '[]', '[1,2,[3,-4]]',
               '{"a":1,"b":{"c":[{"d":"x y"},-7]}}'):
        assert accepts(g, ok), ok
    for bad in ('{', '[1,]', '{"a" 1}', '{"a":}', '[1 2]', '"A"'):
        assert not accepts(g, bad), bad
    assert viable(g, '{"deep":[[[[[')


def test_gbnf_regex_cross_check():
    """Oracle test: grammars expressible in both engines must agree on
    random strings (acceptance AND viability)."""
    import random

    from fusioninfer_amd.guided import GbnfGrammar, RegexGrammar

    pairs = [
        ('root ::= "a" [b-d]* "e"?', "a[b-d]*e?"),
        ('root ::= ("x" | "yz")+', "(x|yz)+"),
        ('root ::= [0-9]{2,4} ("-" [a-c])?', "[0-9]{2,4}(-[a-c])?"),
    ]
    rng = random.Random(7)
    alphabet = "abcdexyz0123456789-"
    for gb_text, rx in pairs:
        gb = GbnfGrammar(gb_text)
        rg = RegexGrammar(rx)
        for _ in range(300):
            s = "".join(rng.choice(alphabet)
                        for _ in range(rng.randrange(0, 8)))
            assert accepts(gb, s) == accepts(rg, s), (gb_text, s)
            assert viable(gb, s) == viable(rg, s), (gb_text, s)
