"""Guided decoding (structured outputs): grammar-constrained sampling.

Capability parity with the reference engines' structured-output surface
(vLLM guided decoding: OpenAI `response_format={"type": "json_object"}`
and the `guided_json` / `guided_regex` / `guided_choice` request
extensions). There is no network access for grammar libraries (outlines /
xgrammar), so the machinery is self-contained:

 * a lazy-DFA regex engine built on Brzozowski derivatives — each state
   IS a (hash-consed) regex AST; stepping a character takes the
   derivative. Supported syntax: literals, `.`, `[...]` classes (ranges,
   negation), escapes (\\d \\w \\s \\D \\W \\S and literal escapes),
   groups `(...)` / `(?:...)`, alternation `|`, quantifiers
   `* + ? {m} {m,} {m,n}`. Anchored fullmatch semantics.
 * a hand-written JSON pushdown grammar for `json_object` mode. Mask
   caching stays finite at any nesting depth because the allowed-char
   set depends only on (mode, extra, stack top, stack empty?).
 * `guided_choice` compiles to an alternation of escaped literals;
   `guided_json` (JSON-schema subset) compiles to a regex.

Per-step cost: the token mask for a grammar state is computed once by
walking the vocabulary trie with the grammar (dead prefixes prune whole
subtrees) and cached per state signature — generation revisits a small
set of DFA states, so steady-state masking is a dict hit + one
masked_fill per guided sequence.
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional, Tuple

import torch

# =============================================================== regex AST
# Nodes are tuples (hashable, so derivative states can be memoized):
#   ('empty',)            matches nothing (dead)
#   ('eps',)              matches the empty string
#   ('cls', frozenset, neg)  one char in (or not in) the set
#   ('cat', r, s) ('alt', r, s) ('star', r)

EMPTY = ("empty",)
EPS = ("eps",)

_D = frozenset("0123456789")
_W = frozenset(
    "abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789_"
)
_S = frozenset(" \t\n\r\f\v")


def _cls(chars, neg=False):
    return ("cls", frozenset(chars), neg)


def _cat(r, s):
    if r == EMPTY or s == EMPTY:
        return EMPTY
    if r == EPS:
        return s
    if s == EPS:
        return r
    return ("cat", r, s)


def _alt(r, s):
    if r == EMPTY:
        return s
    if s == EMPTY:
        return r
    if r == s:
        return r
    return ("alt", r, s)


def _star(r):
    if r in (EMPTY, EPS):
        return EPS
    if r[0] == "star":
        return r
    return ("star", r)


def _nullable(r) -> bool:
    tag = r[0]
    if tag == "eps":
        return True
    if tag in ("empty", "cls"):
        return False
    if tag == "cat":
        return _nullable(r[1]) and _nullable(r[2])
    if tag == "alt":
        return _nullable(r[1]) or _nullable(r[2])
    return True  # star


def _deriv(r, c, memo):
    key = (r, c)
    hit = memo.get(key)
    if hit is not None:
        return hit
    tag = r[0]
    if tag in ("empty", "eps"):
        out = EMPTY
    elif tag == "cls":
        inside = c in r[1]
        out = EPS if (inside != r[2]) else EMPTY
    elif tag == "cat":
        out = _cat(_deriv(r[1], c, memo), r[2])
        if _nullable(r[1]):
            out = _alt(out, _deriv(r[2], c, memo))
    elif tag == "alt":
        out = _alt(_deriv(r[1], c, memo), _deriv(r[2], c, memo))
    else:  # star
        out = _cat(_deriv(r[1], c, memo), r)
    memo[key] = out
    return out


# ---------------------------------------------------------- regex parser
class RegexError(ValueError):
    pass


_ESCAPES = {
    "d": _cls(_D), "D": _cls(_D, True),
    "w": _cls(_W), "W": _cls(_W, True),
    "s": _cls(_S), "S": _cls(_S, True),
    "n": _cls("\n"), "t": _cls("\t"), "r": _cls("\r"),
}


def parse_regex(pattern: str):
    """Recursive-descent parse into the AST above (fullmatch semantics)."""
    pos = [0]
    n = len(pattern)

    def peek():
        return pattern[pos[0]] if pos[0] < n else None

    def take():
        c = pattern[pos[0]]
        pos[0] += 1
        return c

    def parse_alt():
        r = parse_cat()
        while peek() == "|":
            take()
            r = _alt(r, parse_cat())
        return r

    def parse_cat():
        r = EPS
        while peek() is not None and peek() not in "|)":
            r = _cat(r, parse_quant())
        return r

    def parse_quant():
        r = parse_atom()
        while True:
            c = peek()
            if c == "*":
                take()
                r = _star(r)
            elif c == "+":
                take()
                r = _cat(r, _star(r))
            elif c == "?":
                take()
                r = _alt(r, EPS)
            elif c == "{":
                save = pos[0]
                take()
                spec = ""
                while peek() is not None and peek() != "}":
                    spec += take()
                if peek() != "}":
                    pos[0] = save
                    break
                take()
                r = _repeat(r, spec, pattern)
            else:
                break
        return r

    def _repeat(r, spec, pattern):
        parts = spec.split(",")
        try:
            if len(parts) == 1:
                m = x = int(parts[0])
            elif parts[1] == "":
                m, x = int(parts[0]), None
            else:
                m, x = int(parts[0]), int(parts[1])
        except ValueError:
            raise RegexError(f"bad repeat {{{spec}}} in {pattern!r}")
        if x is not None and (x < m or x > 256) or m > 256:
            raise RegexError(f"repeat bound too large in {pattern!r}")
        out = EPS
        for _ in range(m):
            out = _cat(out, r)
        if x is None:
            out = _cat(out, _star(r))
        else:
            opt = _alt(r, EPS)
            for _ in range(x - m):
                out = _cat(out, opt)
        return out

    def parse_atom():
        c = peek()
        if c is None:
            return EPS
        if c == "(":
            take()
            if peek() == "?":
                take()
                if peek() != ":":
                    raise RegexError("only (?:...) groups are supported")
                take()
            r = parse_alt()
            if peek() != ")":
                raise RegexError(f"unbalanced '(' in {pattern!r}")
            take()
            return r
        if c == "[":
            take()
            return parse_class()
        if c == ".":
            take()
            return _cls("\n", True)
        if c == "\\":
            take()
            e = take()
            return _ESCAPES.get(e, _cls(e))
        if c in "*+?{":
            raise RegexError(f"dangling quantifier in {pattern!r}")
        return _cls(take())

    def parse_class():
        neg = False
        if peek() == "^":
            take()
            neg = True
        chars = set()
        first = True
        while True:
            c = peek()
            if c is None:
                raise RegexError(f"unterminated [ in {pattern!r}")
            if c == "]" and not first:
                take()
                break
            first = False
            c = take()
            if c == "\\":
                e = take()
                esc = _ESCAPES.get(e)
                if esc is not None and not esc[2]:
                    chars |= esc[1]
                    continue
                c = {"n": "\n", "t": "\t", "r": "\r"}.get(e, e)
            if peek() == "-" and pos[0] + 1 < n and pattern[pos[0] + 1] != "]":
                take()
                hi = take()
                if hi == "\\":
                    hi = take()
                if ord(hi) < ord(c):
                    raise RegexError(f"bad range {c}-{hi} in {pattern!r}")
                chars |= {chr(x) for x in range(ord(c), ord(hi) + 1)}
            else:
                chars.add(c)
        return _cls(chars, neg)

    r = parse_alt()
    if pos[0] != n:
        raise RegexError(f"trailing {pattern[pos[0]:]!r} in {pattern!r}")
    return r


# ============================================================== grammars
class RegexGrammar:
    """Lazy DFA over Brzozowski derivatives; a state is a regex AST."""

    def __init__(self, pattern: str):
        self.pattern = pattern
        self.root = parse_regex(pattern)
        self._memo: Dict[Tuple, object] = {}

    def initial(self):
        return self.root

    def step(self, state, ch):
        out = _deriv(state, ch, self._memo)
        return None if out == EMPTY else out

    def is_complete(self, state) -> bool:
        return _nullable(state)

    def can_extend(self, state) -> bool:
        return state != EPS

    def signature(self, state):
        return state


class JsonGrammar:
    """Pushdown grammar for well-formed JSON. State = (mode, extra, stack);
    allowed characters depend only on (mode, extra, stack top, empty?), so
    mask caching is finite at any depth. Compact form (no optional
    whitespace at the top level, so termination is decidable; whitespace
    IS allowed at structural positions inside containers).

    root_object=True (OpenAI json_object mode) requires the value to be an
    object; False accepts any JSON value.
    """

    WS = " \t\n\r"

    def __init__(self, root_object: bool = True):
        self.root_object = root_object

    def initial(self):
        return ("val", "root" if self.root_object else None, ())

    # ------------------------------------------------------------- helpers
    def _post(self, stack):
        return ("post", None, stack)

    def _post_step(self, stack, ch):
        """Transitions out of 'a value just ended' — also used by number
        states when a delimiter arrives."""
        if not stack:
            return None  # root value done: nothing may follow
        if ch in self.WS:
            return self._post(stack)
        top = stack[-1]
        if top == "{":
            if ch == ",":
                return ("key", None, stack)
            if ch == "}":
                return self._post(stack[:-1])
        else:  # '['
            if ch == ",":
                return ("val", None, stack)
            if ch == "]":
                return self._post(stack[:-1])
        return None

    # ---------------------------------------------------------------- step
    def step(self, state, ch):
        mode, extra, stack = state
        if mode == "val":
            if ch in self.WS:
                return state
            if extra == "close" and ch == "]":
                return self._post(stack[:-1])  # empty array
            if extra == "root" and ch != "{":
                return None  # json_object mode: root must be an object
            if ch == "{":
                return ("key", "first", stack + ("{",))
            if ch == "[":
                return ("val", "close", stack + ("[",))
            if ch == '"':
                return ("str", "val", stack)
            if ch == "-":
                return ("num", "sign", stack)
            if ch == "0":
                return ("num", "int0", stack)
            if ch in "123456789":
                return ("num", "int", stack)
            if ch == "t":
                return ("lit", "rue", stack)
            if ch == "f":
                return ("lit", "alse", stack)
            if ch == "n":
                return ("lit", "ull", stack)
            return None
        if mode == "key":
            if ch in self.WS:
                return (mode, extra, stack)
            if ch == '"':
                return ("str", "key", stack)
            if extra == "first" and ch == "}":
                return self._post(stack[:-1])  # empty object
            return None
        if mode == "colon":
            if ch in self.WS:
                return state
            if ch == ":":
                return ("val", None, stack)
            return None
        if mode == "str":
            if ch == '"':
                if extra == "key":
                    return ("colon", None, stack)
                return self._post(stack)
            if ch == "\\":
                return ("esc", extra, stack)
            if ch in "\n\r":
                return None
            return state
        if mode == "esc":
            if ch == "u":
                return ("hex", (extra, 4), stack)
            if ch in '"\\/bfnrt':
                return ("str", extra, stack)
            return None
        if mode == "hex":
            ctx, left = extra
            if ch in "0123456789abcdefABCDEF":
                return ("str", ctx, stack) if left == 1 else \
                    ("hex", (ctx, left - 1), stack)
            return None
        if mode == "lit":
            if extra and ch == extra[0]:
                rest = extra[1:]
                return ("lit", rest, stack) if rest else self._post(stack)
            return None
        if mode == "num":
            sub = extra
            if sub == "sign":
                if ch == "0":
                    return ("num", "int0", stack)
                if ch in "123456789":
                    return ("num", "int", stack)
                return None
            accepting = sub in ("int", "int0", "frac", "exp")
            if sub in ("int", "exp") and ch in _D:
                return state
            # strict JSON: no digits after a leading 0 — int0 digits fall
            # through to _post_step below, which rejects them
            if sub in ("int", "int0") and ch == ".":
                return ("num", "frac0", stack)
            if sub in ("int", "int0", "frac") and ch in "eE":
                return ("num", "e0", stack)
            if sub in ("frac0", "frac") and ch in _D:
                return ("num", "frac", stack)
            if sub == "e0":
                if ch in "+-":
                    return ("num", "esign", stack)
                if ch in _D:
                    return ("num", "exp", stack)
                return None
            if sub == "esign" and ch in _D:
                return ("num", "exp", stack)
            if accepting:
                return self._post_step(stack, ch)
            return None
        # mode == "post"
        return self._post_step(stack, ch)

    def is_complete(self, state) -> bool:
        mode, extra, stack = state
        if stack:
            return False
        if mode == "post":
            return True
        return mode == "num" and extra in ("int", "int0", "frac", "exp")

    def can_extend(self, state) -> bool:
        mode, extra, stack = state
        if mode == "post" and not stack:
            return False
        return True

    def signature(self, state):
        mode, extra, stack = state
        return (mode, extra, stack[-1] if stack else None, bool(stack))


# ==================================================== schema -> regex
_RE_SPECIAL = set(".^$*+?{}[]\\|()")


def _re_escape(s: str) -> str:
    return "".join("\\" + c if c in _RE_SPECIAL else c for c in s)


_STRING_RE = '"([^"\\\\\n\r]|\\\\.)*"'
_INT_RE = "-?(0|[1-9][0-9]*)"
_NUM_RE = "-?(0|[1-9][0-9]*)(\\.[0-9]+)?([eE][+-]?[0-9]+)?"


def _int_range_regex(lo, hi) -> str:
    """Regex for integers in [lo, hi] (xgrammar-class bounds support).
    Small ranges enumerate exactly; unbounded non-negative strips the
    sign; anything else is unsupported (caller falls back)."""
    if lo is not None and hi is not None:
        lo, hi = int(lo), int(hi)
        if hi < lo:
            raise ValueError("maximum < minimum")
        if hi - lo <= 512:
            return "(" + "|".join(str(v) for v in range(lo, hi + 1)) + ")"
        raise ValueError("integer range too wide to enumerate")
    if lo is not None and int(lo) >= 0:
        return "(0|[1-9][0-9]*)" if int(lo) == 0 else "[1-9][0-9]*"
    if hi is not None and int(hi) <= 0:
        return "(-[1-9][0-9]*|0)" if int(hi) == 0 else "-[1-9][0-9]*"
    raise ValueError("unsupported integer bounds")


def schema_to_regex(schema: dict, depth: int = 0, defs=None) -> str:
    """JSON-schema subset -> anchored regex (compact output, property
    order fixed; properties outside `required` are optional — emitted
    in order or skipped). Supported beyond the basics: enum/const,
    anyOf/oneOf, allOf (single branch), type lists, string pattern +
    minLength/maxLength, integer minimum/maximum (enumerable or
    sign-determined), array minItems/maxItems, local $defs/$ref
    (cycle-guarded by depth). Unsupported constructs raise ValueError —
    the API layer then falls back to generic JSON."""
    if depth > 8:
        raise ValueError("schema nesting too deep")
    if not isinstance(schema, dict):
        raise ValueError("schema must be an object")
    if defs is None:
        defs = schema.get("$defs") or schema.get("definitions") or {}
    if "$ref" in schema:
        ref = schema["$ref"]
        for prefix in ("#/$defs/", "#/definitions/"):
            if ref.startswith(prefix):
                name = ref[len(prefix):]
                if name not in defs:
                    raise ValueError(f"unresolved $ref {ref!r}")
                return schema_to_regex(defs[name], depth + 1, defs)
        raise ValueError(f"unsupported $ref {ref!r} (local #/$defs only)")
    if "allOf" in schema:
        branches = schema["allOf"]
        if len(branches) != 1:
            raise ValueError("allOf with multiple branches unsupported")
        merged = {**branches[0],
                  **{k: v for k, v in schema.items() if k != "allOf"}}
        return schema_to_regex(merged, depth + 1, defs)
    if "enum" in schema:
        opts = "|".join(
            _re_escape(json.dumps(v, separators=(",", ":")))
            for v in schema["enum"]
        )
        return f"({opts})"
    if "const" in schema:
        return _re_escape(json.dumps(schema["const"], separators=(",", ":")))
    for alt_key in ("anyOf", "oneOf"):
        if alt_key in schema:
            opts = "|".join(
                schema_to_regex(s, depth + 1, defs) for s in schema[alt_key]
            )
            return f"({opts})"
    t = schema.get("type")
    if isinstance(t, list):  # {"type": ["string", "null"]}
        opts = "|".join(
            schema_to_regex({**schema, "type": tt}, depth + 1, defs)
            for tt in t
        )
        return f"({opts})"
    if t == "string":
        pattern = schema.get("pattern")
        if pattern:
            body = pattern.lstrip("^").rstrip("$")
            parse_regex(body)  # unsupported syntax raises -> fallback
            return f'"({body})"'
        lo = schema.get("minLength")
        hi = schema.get("maxLength")
        if lo is not None or hi is not None:
            lo = int(lo or 0)
            if hi is not None and (int(hi) < lo or int(hi) > 4096):
                raise ValueError("bad minLength/maxLength")
            rep = f"{{{lo},{int(hi)}}}" if hi is not None else f"{{{lo},}}"
            return f'"([^"\\\n\r]|\\.){rep}"'
        return _STRING_RE
    if t == "integer":
        if "minimum" in schema or "maximum" in schema:
            return _int_range_regex(schema.get("minimum"),
                                    schema.get("maximum"))
        return _INT_RE
    if t == "number":
        return _NUM_RE
    if t == "boolean":
        return "(true|false)"
    if t == "null":
        return "null"
    if t == "array":
        item = schema_to_regex(schema.get("items", {"type": "string"}),
                               depth + 1, defs)
        lo = int(schema.get("minItems", 0))
        hi = schema.get("maxItems")
        if lo == 0 and hi is None:
            return f"\\[({item}(,{item})*)?\\]"
        if hi is not None and (int(hi) < lo or int(hi) > 64):
            raise ValueError("bad minItems/maxItems")
        tail = (f"(,{item}){{{max(lo - 1, 0)},{int(hi) - 1}}}"
                if hi is not None else f"(,{item}){{{max(lo - 1, 0)},}}")
        body = f"{item}{tail}"
        return f"\\[{body}\\]" if lo > 0 else f"\\[({body})?\\]"
    if t == "object":
        props = schema.get("properties") or {}
        if not props:
            raise ValueError("object schema needs properties")
        # JSON-schema semantics: properties outside `required` are
        # optional. Deviation (documented): with NO `required` key every
        # listed property is emitted — the useful default for extraction
        # prompts, and the historical behavior. Property order is fixed
        # to the schema's listing order either way.
        req_list = schema.get("required")
        required = set(props.keys() if req_list is None else req_list)
        rendered = [
            (f"{_re_escape(json.dumps(name))}:"
             f"{schema_to_regex(sub, depth + 1, defs)}",
             name in required)
            for name, sub in props.items()
        ]
        if sum(1 for _, req in rendered if not req) > 12:
            raise ValueError("too many optional properties")
        # tails[i]: properties i.. when some property was already
        # emitted (each present one is comma-prefixed; optional ones
        # are skippable groups)
        tails = [""]
        for part, req in reversed(rendered):
            tails.append(f"(,{part}){'' if req else '?'}{tails[-1]}")
        tails.reverse()
        # head: alternation over which property appears FIRST (no
        # comma). Only properties up to and including the first
        # required one can be first.
        k = next((i for i, (_, req) in enumerate(rendered) if req),
                 len(rendered))
        starts = [f"{part}{tails[i + 1]}"
                  for i, (part, _) in enumerate(rendered[:k + 1])]
        if k == len(rendered):  # no required property: {} is valid too
            body = "(" + "|".join(starts) + ")?"
        elif len(starts) == 1:
            body = starts[0]
        else:
            body = "(" + "|".join(starts) + ")"
        return "\\{" + body + "\\}"
    raise ValueError(f"unsupported schema type {t!r}")


# ============================================================ vocabulary
class Vocabulary:
    """Token id -> decoded string, plus a trie over unique strings so a
    grammar state's token mask is computed by pruned DFS instead of a
    per-token scan."""

    def __init__(self, vocab_size: int, decode_one):
        self.vocab_size = vocab_size
        by_string: Dict[str, List[int]] = {}
        self.strings: List[str] = []
        for t in range(vocab_size):
            s = decode_one(t)
            self.strings.append(s)
            if s:  # tokens decoding to "" can never be grammar-checked
                by_string.setdefault(s, []).append(t)
        # trie node: (children: {ch: node}, ids ending here)
        self.trie = ({}, [])
        for s, ids in by_string.items():
            node = self.trie
            for ch in s:
                node = node[0].setdefault(ch, ({}, []))
            node[1].extend(ids)


class GuidedMaskCache:
    """Per-(grammar, vocab, device) cache of token masks by state sig."""

    def __init__(self, grammar, vocab: Vocabulary):
        self.grammar = grammar
        self.vocab = vocab
        self._masks: Dict[Tuple, torch.Tensor] = {}

    def mask(self, state, device) -> torch.Tensor:
        key = (self.grammar.signature(state), str(device))
        m = self._masks.get(key)
        if m is not None:
            return m
        allowed: List[int] = []
        g = self.grammar

        def dfs(node, st):
            children, ids = node
            if ids:
                allowed.extend(ids)
            for ch, child in children.items():
                nxt = g.step(st, ch)
                if nxt is not None:
                    dfs(child, nxt)

        # root children only: step once per distinct first char
        for ch, child in self.vocab.trie[0].items():
            nxt = g.step(state, ch)
            if nxt is not None:
                dfs(child, nxt)
        m = torch.zeros(self.vocab.vocab_size, dtype=torch.bool)
        if allowed:
            m[torch.tensor(allowed, dtype=torch.long)] = True
        m = m.to(device)
        self._masks[key] = m
        return m


# ============================================================ GBNF grammar
class _GbnfParser:
    """GBNF (llama.cpp grammar format) parser. Produces
    {rule: [alt, ...]} where alt is a tuple of symbols and a symbol is
    ("t", ranges, negated) — a char-class terminal with ranges a tuple
    of (lo, hi) ordinals — or ("r", name). Groups and repetition
    suffixes desugar into synthetic right-recursive rules (so * + ?
    never introduce left recursion)."""

    def __init__(self, text: str):
        self.s = text
        self.i = 0
        self.rules: Dict[str, list] = {}
        self._gen = 0

    def err(self, msg: str):
        line = self.s.count("\n", 0, self.i) + 1
        raise ValueError(f"GBNF parse error (line {line}): {msg}")

    def ws(self):
        while self.i < len(self.s):
            c = self.s[self.i]
            if c == "#":  # comment to end of line
                while self.i < len(self.s) and self.s[self.i] != "\n":
                    self.i += 1
            elif c in " \t\r\n":
                self.i += 1
            else:
                return

    def _at_rule_start(self) -> bool:
        j = self.i
        while j < len(self.s) and (self.s[j].isalnum()
                                   or self.s[j] in "_-"):
            j += 1
        if j == self.i:
            return False
        while j < len(self.s) and self.s[j] in " \t":
            j += 1
        return self.s[j:j + 3] == "::=" or self.s[j:j + 1] == ":"

    def ident(self) -> str:
        j = self.i
        while j < len(self.s) and (self.s[j].isalnum() or self.s[j] in "_-"):
            j += 1
        if j == self.i:
            self.err("expected rule name")
        name, self.i = self.s[self.i:j], j
        return name

    def parse(self) -> Dict[str, list]:
        while True:
            self.ws()
            if self.i >= len(self.s):
                break
            name = self.ident()
            self.ws()
            if self.s.startswith("::=", self.i):
                self.i += 3
            elif self.s.startswith(":", self.i):
                self.i += 1
            else:
                self.err(f"expected '::=' after {name!r}")
            self.rules[name] = self.alternates()
        return self.rules

    def fresh(self, alts) -> tuple:
        self._gen += 1
        name = f"%g{self._gen}"
        self.rules[name] = alts
        return ("r", name)

    def alternates(self) -> list:
        alts = [self.sequence()]
        while True:
            self.ws()
            if self.i < len(self.s) and self.s[self.i] == "|":
                self.i += 1
                alts.append(self.sequence())
            else:
                return alts

    def sequence(self) -> tuple:
        syms: list = []
        while True:
            self.ws()
            if self.i >= len(self.s):
                return tuple(syms)
            c = self.s[self.i]
            if c in "|)" or self._at_rule_start():
                return tuple(syms)
            item = self.item()  # list of symbols
            self.ws()
            rep = self.s[self.i] if self.i < len(self.s) else ""
            lo = hi = None
            if rep and rep in "*+?":
                self.i += 1
                lo, hi = {"*": (0, None), "+": (1, None),
                          "?": (0, 1)}[rep]
            elif rep == "{":
                j = self.s.index("}", self.i)
                body = self.s[self.i + 1:j]
                self.i = j + 1
                parts = body.split(",")
                lo = int(parts[0])
                hi = (lo if len(parts) == 1 else
                      (None if parts[1].strip() == "" else int(parts[1])))
                if hi is not None and (hi < lo or hi > 64):
                    self.err("bad {m,n} bounds")
            if lo is None:
                syms.extend(item)
                continue
            unit = (item[0] if len(item) == 1
                    else self.fresh([tuple(item)]))
            out = [unit] * lo
            if hi is None:  # X{lo,} -> lo copies + X*
                star = self.fresh([])
                self.rules[star[1]] = [(unit, star), ()]
                out.append(star)
            else:
                opt = self.fresh([(unit,), ()])
                out.extend([opt] * (hi - lo))
            syms.extend(out)

    def item(self) -> list:
        c = self.s[self.i]
        if c == '"':
            return [("t", ((o, o),), False) for o in self.literal()]
        if c == "[":
            return [self.char_class()]
        if c == "(":
            self.i += 1
            alts = self.alternates()
            self.ws()
            if self.i >= len(self.s) or self.s[self.i] != ")":
                self.err("expected ')'")
            self.i += 1
            return [self.fresh(alts)]
        if c.isalnum() or c in "_-":
            return [("r", self.ident())]
        self.err(f"unexpected {c!r}")

    def _escape(self) -> int:
        c = self.s[self.i]
        self.i += 1
        if c != "\\":
            return ord(c)
        e = self.s[self.i]
        self.i += 1
        if e == "x":
            v = int(self.s[self.i:self.i + 2], 16)
            self.i += 2
            return v
        if e == "u":
            v = int(self.s[self.i:self.i + 4], 16)
            self.i += 4
            return v
        return ord({"n": "\n", "t": "\t", "r": "\r"}.get(e, e))

    def literal(self) -> list:
        self.i += 1  # opening quote
        out = []
        while self.i < len(self.s) and self.s[self.i] != '"':
            out.append(self._escape())
        if self.i >= len(self.s):
            self.err("unterminated string")
        self.i += 1
        if not out:
            self.err("empty literal")
        return out

    def char_class(self):
        self.i += 1  # [
        neg = self.s[self.i] == "^"
        if neg:
            self.i += 1
        ranges = []
        while self.i < len(self.s) and self.s[self.i] != "]":
            lo = self._escape()
            if (self.s[self.i] == "-" and self.i + 1 < len(self.s)
                    and self.s[self.i + 1] != "]"):
                self.i += 1
                hi = self._escape()
            else:
                hi = lo
            ranges.append((lo, hi))
        if self.i >= len(self.s):
            self.err("unterminated char class")
        self.i += 1
        if not ranges:
            self.err("empty char class")
        return ("t", tuple(ranges), neg)


class GbnfGrammar:
    """GBNF (llama.cpp grammar format) engine — vLLM guided_grammar
    parity. State = frozenset of nondeterministic parse stacks
    (llama.cpp's scheme): a stack is a tuple of frames (alt, pos) with
    the top frame last, normalized so the top's current symbol is a
    terminal; the empty stack () means the root completed. Left
    recursion is rejected (expansion-depth bound), the same limitation
    llama.cpp documents."""

    _MAX_STACKS = 4096

    def __init__(self, text: str, root: str = "root"):
        self.rules = _GbnfParser(text).parse()
        if root not in self.rules:
            raise ValueError(f"GBNF grammar has no {root!r} rule")
        for alts in self.rules.values():
            for alt in alts:
                for sym in alt:
                    if sym[0] == "r" and sym[1] not in self.rules:
                        raise ValueError(f"undefined rule {sym[1]!r}")
        out: set = set()
        self._norm((((("r", root),), 0),), out)
        self._initial = frozenset(out)

    def _norm(self, stack, out, depth=0):
        if depth > 256:
            raise ValueError(
                "GBNF expansion too deep (left recursion is unsupported)"
            )
        while True:
            if not stack:
                out.add(())
                return
            alt, pos = stack[-1]
            if pos >= len(alt):
                stack = stack[:-1]
                if stack:
                    palt, ppos = stack[-1]
                    stack = stack[:-1] + ((palt, ppos + 1),)
                continue
            sym = alt[pos]
            if sym[0] == "t":
                out.add(stack)
                return
            # tail-call elision: a ref in FINAL position drops its frame
            # before expanding (the child's completion continues with the
            # grandparent directly). Without this, right-recursive
            # desugared repetitions (X* -> R := X R | eps) would grow the
            # stack by one frame per consumed repetition and trip the
            # depth guard after ~256 characters.
            base = stack[:-1] if pos + 1 == len(alt) else stack
            for a in self.rules[sym[1]]:
                self._norm(base + ((a, 0),), out, depth + 1)
            return

    def initial(self):
        return self._initial

    def step(self, state, ch):
        o = ord(ch)
        nxt: set = set()
        for stack in state:
            if not stack:
                continue  # completed root: nothing may follow
            alt, pos = stack[-1]
            _, ranges, neg = alt[pos]
            hit = any(lo <= o <= hi for lo, hi in ranges)
            if hit != neg:
                self._norm(stack[:-1] + ((alt, pos + 1),), nxt)
        if not nxt:
            return None
        if len(nxt) > self._MAX_STACKS:
            raise ValueError("GBNF state explosion")
        return frozenset(nxt)

    def is_complete(self, state) -> bool:
        return () in state

    def can_extend(self, state) -> bool:
        return any(stack for stack in state)

    def signature(self, state):
        return state


class GuidedDecoder:
    """Per-request grammar cursor the engine consults at sampling time.

    allowed_mask() -> bool[V] (None = no token fits: the engine finishes
    the request — cleanly when the grammar is complete).
    advance_token() steps the grammar over the emitted token's text;
    is_terminal() tells the engine to stop (grammar complete and not
    extendable)."""

    def __init__(self, cache: GuidedMaskCache):
        self.cache = cache
        self.state = cache.grammar.initial()
        self.dead = False

    def allowed_mask(self, device) -> Optional[torch.Tensor]:
        if self.dead:
            return None
        m = self.cache.mask(self.state, device)
        return m if bool(m.any()) else None

    def advance_token(self, token_id: int) -> None:
        g = self.cache.grammar
        s = self.state
        for ch in self.cache.vocab.strings[token_id]:
            s = g.step(s, ch)
            if s is None:
                self.dead = True
                return
        self.state = s

    def is_terminal(self) -> bool:
        if self.dead:
            return True
        g = self.cache.grammar
        return g.is_complete(self.state) and not g.can_extend(self.state)


# ============================================================== factory
_CACHES: Dict[Tuple, GuidedMaskCache] = {}


def _byteize(text: str) -> str:
    """UTF-8 bytes of `text` as a latin-1 string (ASCII fixed-point).
    Grammar specs are matched against BYTE-level token strings
    (Vocabulary is built from decode_one_bytes), so non-ASCII literal
    chars in user regex/choice/GBNF specs must expand to their UTF-8
    byte sequence. Limitation: character-class RANGES with non-ASCII
    endpoints are not byte-expandable and will mismatch."""
    return text.encode("utf-8").decode("latin-1")


def build_guided(kind: str, spec, vocab: Vocabulary) -> GuidedDecoder:
    """kind: 'json_object' | 'regex' | 'choice' | 'json_schema'.
    Raises ValueError on unsupported specs (API layer maps it to 400)."""
    if kind == "json_object":
        key = ("json_object", id(vocab))
        if key not in _CACHES:
            _CACHES[key] = GuidedMaskCache(JsonGrammar(root_object=True),
                                           vocab)
        return GuidedDecoder(_CACHES[key])
    if kind == "regex":
        spec = _byteize(spec)
        key = ("regex", spec, id(vocab))
        if key not in _CACHES:
            _CACHES[key] = GuidedMaskCache(RegexGrammar(spec), vocab)
        return GuidedDecoder(_CACHES[key])
    if kind == "choice":
        pattern = "(" + "|".join(_re_escape(str(c)) for c in spec) + ")"
        return build_guided("regex", pattern, vocab)
    if kind == "json_schema":
        try:
            pattern = schema_to_regex(spec)
        except ValueError:
            # unsupported schema construct: enforce well-formed JSON
            return build_guided("json_object", vocab=vocab, spec=None)
        return build_guided("regex", pattern, vocab)
    if kind == "grammar":
        spec = _byteize(spec)
        key = ("grammar", spec, id(vocab))
        if key not in _CACHES:
            _CACHES[key] = GuidedMaskCache(GbnfGrammar(spec), vocab)
        return GuidedDecoder(_CACHES[key])
    raise ValueError(f"unknown guided decoding kind {kind!r}")
