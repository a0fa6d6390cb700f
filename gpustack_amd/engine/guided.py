"""Grammar-constrained JSON decoding (reference: vLLM guided_json via
outlines/xgrammar, passed through gpustack backend_parameters).

First-party design: a character-level machine validates the generated text
incrementally; sampling picks the highest-probability token whose decoded
string keeps the output a valid prefix (top-k candidate rejection — the
top few tokens are almost always valid, so the per-step cost is a handful
of short string walks, not a vocab scan).

Two grammars:
  * `guided_json: true` — any syntactically valid JSON value (full PDA:
    nested objects/arrays, strings with escapes, numbers, literals).
  * `guided_json: {schema}` — a canonical template compiled from a JSON
    Schema subset (flat or nested objects with string/number/integer/
    boolean/null/enum/array-typed properties, all emitted in property
    order): {"name":<string>,"age":<integer>} — literal segments are
    forced, value slots run the matching value machine.

Machines implement `advance(ch) -> bool` (consume or reject) and
`.complete` (the text so far is a finished value). Sequencing retries a
rejected char on the next segment when the current one is complete.
"""
from __future__ import annotations

import copy

_WS = " \t\n\r"
_DIGITS = "0123456789"
_HEX = "0123456789abcdefABCDEF"


class StringM:
    """A JSON string including both quotes. Supports the JSON-Schema
    maxLength/minLength bounds (counted in raw body characters, so escape
    sequences count conservatively high — the cap still holds)."""

    def __init__(self, max_len: int | None = None, min_len: int = 0):
        self.state = 0  # 0=expect open ", 1=body, 2=escape, 3..6=\uXXXX, 7=done
        self.max_len = max_len
        self.min_len = min_len
        self.n = 0

    @property
    def complete(self) -> bool:
        return self.state == 7

    def advance(self, ch: str) -> bool:
        s = self.state
        if s == 0:
            if ch == '"':
                self.state = 1
                return True
            return False
        if s == 1:
            if ch == '"':
                if self.n < self.min_len:
                    return False
                self.state = 7
                return True
            if self.max_len is not None and self.n >= self.max_len:
                return False  # only the closing quote may follow
            self.n += 1
            if ch == "\\":
                self.state = 2
                return True
            return ord(ch) >= 0x20
        if s == 2:
            if ch in '"\\/bfnrt':
                self.state = 1
                return True
            if ch == "u":
                self.state = 3
                return True
            return False
        if 3 <= s <= 6:
            if ch in _HEX:
                self.state = 1 if s == 6 else s + 1
                return True
            return False
        return False  # done: no more chars


class NumberM:
    """-?(0|[1-9][0-9]*)(\\.[0-9]+)?([eE][+-]?[0-9]+)?; integer_only drops
    the fraction/exponent parts."""

    def __init__(self, integer_only: bool = False):
        self.integer_only = integer_only
        self.state = 0  # 0=start 1=after- 2=after0 3=int 4=dot 5=frac 6=e 7=esign 8=exp

    @property
    def complete(self) -> bool:
        return self.state in (2, 3, 5, 8)

    def advance(self, ch: str) -> bool:
        s = self.state
        if s == 0:
            if ch == "-":
                self.state = 1
                return True
            if ch == "0":
                self.state = 2
                return True
            if ch in "123456789":
                self.state = 3
                return True
            return False
        if s == 1:
            if ch == "0":
                self.state = 2
                return True
            if ch in "123456789":
                self.state = 3
                return True
            return False
        if s in (2, 3):
            if s == 3 and ch in _DIGITS:
                return True
            if self.integer_only:
                return False
            if ch == ".":
                self.state = 4
                return True
            if ch in "eE":
                self.state = 6
                return True
            return False
        if s in (4, 5):
            if ch in _DIGITS:
                self.state = 5
                return True
            if s == 5 and ch in "eE":
                self.state = 6
                return True
            return False
        if s == 6:
            if ch in "+-":
                self.state = 7
                return True
            if ch in _DIGITS:
                self.state = 8
                return True
            return False
        if s == 7:
            if ch in _DIGITS:
                self.state = 8
                return True
            return False
        if s == 8:
            return ch in _DIGITS
        return False


class LitM:
    """One of several fixed strings (true/false, null, enum values)."""

    def __init__(self, options: list[str]):
        self.options = options
        self.pos = 0
        self.alive = list(range(len(options)))

    @property
    def complete(self) -> bool:
        return any(len(self.options[i]) == self.pos for i in self.alive)

    def advance(self, ch: str) -> bool:
        nxt = [i for i in self.alive
               if self.pos < len(self.options[i]) and self.options[i][self.pos] == ch]
        if not nxt:
            return False
        self.alive = nxt
        self.pos += 1
        return True


class JsonPDA:
    """Any syntactically valid JSON value (single top-level value)."""

    def __init__(self):
        self.stack: list[str] = []   # 'o' object, 'a' array
        self.mode = "value"          # value|key|colon|comma|string|number|lit
        self.sub = None              # active StringM/NumberM/LitM
        self.sub_is_key = False
        self.done_value = False      # a complete top-level value was consumed

    @property
    def complete(self) -> bool:
        if self.stack:
            return False
        if self.mode == "number" and self.sub is not None:
            return self.sub.complete
        return self.done_value

    def _finish_value(self) -> None:
        self.sub = None
        if not self.stack:
            self.mode = "end"
            self.done_value = True
        else:
            self.mode = "comma"

    def advance(self, ch: str) -> bool:
        m = self.mode
        if m in ("string", "lit"):
            if self.sub.advance(ch):
                if self.sub.complete and m == "string" and self.sub_is_key:
                    if self.sub.state == 7:
                        self.mode = "colon"
                        self.sub = None
                elif self.sub.complete:
                    self._finish_value()
                return True
            return False
        if m == "number":
            if self.sub.advance(ch):
                return True
            if self.sub.complete:  # number ended; re-dispatch ch
                self._finish_value()
                return self.advance(ch)
            return False
        if ch in _WS:
            return m in ("value", "key", "colon", "comma", "end")
        if m == "value":
            if ch == "{":
                self.stack.append("o")
                self.mode = "key"
                return True
            if ch == "[":
                self.stack.append("a")
                self.mode = "value"
                self.first_in_container = True
                return True
            if ch == "]" and self.stack and self.stack[-1] == "a" \
                    and getattr(self, "first_in_container", False):
                self.stack.pop()
                self.first_in_container = False
                self._finish_value()
                return True
            if ch == '"':
                self.sub = StringM()
                self.sub.advance('"')
                self.sub_is_key = False
                self.mode = "string"
                return True
            if ch in "-0123456789":
                self.sub = NumberM()
                self.mode = "number"
                return self.sub.advance(ch)
            if ch in "tfn":
                self.sub = LitM(["true", "false", "null"])
                self.mode = "lit"
                return self.sub.advance(ch)
            return False
        if m == "key":
            if ch == '"':
                self.sub = StringM()
                self.sub.advance('"')
                self.sub_is_key = True
                self.mode = "string"
                return True
            if ch == "}" and self.stack and self.stack[-1] == "o" \
                    and getattr(self, "first_in_container", True):
                self.stack.pop()
                self._finish_value()
                return True
            return False
        if m == "colon":
            if ch == ":":
                self.mode = "value"
                self.first_in_container = False
                return True
            return False
        if m == "comma":
            top = self.stack[-1] if self.stack else None
            if ch == "," and top == "o":
                self.mode = "key"
                self.first_in_container = False
                return True
            if ch == "," and top == "a":
                self.mode = "value"
                self.first_in_container = False
                return True
            if ch == "}" and top == "o":
                self.stack.pop()
                self._finish_value()
                return True
            if ch == "]" and top == "a":
                self.stack.pop()
                self._finish_value()
                return True
            return False
        return False  # end: nothing more


class SeqM:
    """Sequence of machines/literals; rejected chars retry on the next
    segment once the current one is complete."""

    def __init__(self, segments: list):
        self.segments = segments
        self.idx = 0

    @property
    def complete(self) -> bool:
        i = self.idx
        if i >= len(self.segments):
            return True
        # complete if every remaining segment is already complete (only the
        # current can hold partial state)
        return self.segments[i].complete and i == len(self.segments) - 1

    def advance(self, ch: str) -> bool:
        while self.idx < len(self.segments):
            seg = self.segments[self.idx]
            if seg.advance(ch):
                return True
            if seg.complete:
                self.idx += 1
                continue
            return False
        return False


class ArrayM:
    """[item(,item)*] with typed items from a factory."""

    def __init__(self, item_factory):
        self.factory = item_factory
        self.state = 0  # 0=expect [ 1=first item or ] 2=in item 3=comma or ]
        self.item = None

    @property
    def complete(self) -> bool:
        return self.state == 4

    def advance(self, ch: str) -> bool:
        if self.state == 0:
            if ch == "[":
                self.state = 1
                return True
            return False
        if self.state == 1:
            if ch == "]":
                self.state = 4
                return True
            self.item = self.factory()
            self.state = 2
            return self.advance(ch)
        if self.state == 2:
            if self.item.advance(ch):
                return True
            # item can take no more; if it is complete the char must be the
            # delimiter — uniform handling so ambiguous items (numbers,
            # prefix-overlapping enums like "a"/"ab") can keep extending
            if self.item.complete:
                self.state = 3
                return self.advance(ch)
            return False
        if self.state == 3:
            if ch == ",":
                self.item = self.factory()
                self.state = 2
                return True
            if ch == "]":
                self.state = 4
                return True
            return False
        return False


def _value_machine(schema: dict):
    t = schema.get("type")
    if "enum" in schema:
        import json as _json

        return lambda: LitM([_json.dumps(v) for v in schema["enum"]])
    if t == "string":
        mx, mn = schema.get("maxLength"), schema.get("minLength", 0)
        return lambda: StringM(mx, mn)
    if t == "integer":
        return lambda: NumberM(integer_only=True)
    if t == "number":
        return NumberM
    if t == "boolean":
        return lambda: LitM(["true", "false"])
    if t == "null":
        return lambda: LitM(["null"])
    if t == "array":
        inner = _value_machine(schema.get("items", {}))
        return lambda: ArrayM(inner)
    if t == "object" and schema.get("properties"):
        return lambda: compile_schema(schema)
    return JsonPDA  # unconstrained value


class _Lit:
    """Exact literal text segment."""

    def __init__(self, text: str):
        self.text = text
        self.pos = 0

    @property
    def complete(self) -> bool:
        return self.pos >= len(self.text)

    def advance(self, ch: str) -> bool:
        if self.pos < len(self.text) and self.text[self.pos] == ch:
            self.pos += 1
            return True
        return False


def compile_schema(schema: dict) -> SeqM:
    """JSON Schema (object subset) -> canonical template machine.
    Properties are emitted in declaration order, all of them, unquoted
    whitespace-free: {"a":<v>,"b":<v>}."""
    import json as _json

    props = schema.get("properties") or {}
    segs: list = [_Lit("{")]
    for i, (key, sub) in enumerate(props.items()):
        if i:
            segs.append(_Lit(","))
        segs.append(_Lit(_json.dumps(key) + ":"))
        segs.append(_value_machine(sub)())
    segs.append(_Lit("}"))
    return SeqM(segs)


class GuidedJsonState:
    """Per-sequence guided-decoding state: the machine + commit/probe API."""

    def __init__(self, schema):
        if isinstance(schema, dict) and schema.get("type") == "object" \
                and schema.get("properties"):
            self.machine = compile_schema(schema)
        elif isinstance(schema, dict) and (schema.get("type")
                                           or "enum" in schema):
            self.machine = _value_machine(schema)()
        else:
            self.machine = JsonPDA()

    @property
    def complete(self) -> bool:
        return self.machine.complete

    def try_advance(self, text: str):
        """Probe: returns the advanced machine if `text` keeps the output a
        valid prefix (None otherwise); caller commits via `commit`."""
        m = copy.deepcopy(self.machine)
        for ch in text:
            if not m.advance(ch):
                return None
        return m

    def commit(self, machine) -> None:
        self.machine = machine


# ---- guided_regex: regex subset -> NFA with incremental prefix matching ----
# (reference surface: vLLM guided_regex). Supported: literals, escapes
# (\d \w \s \D \W \S and escaped metachars), '.', character classes
# [a-z0-9_] incl. negation, groups, alternation, and the quantifiers
# * + ? {m} {m,} {m,n}. Thompson construction; the machine keeps the
# current state set, so advance() is O(states) per character and
# `complete` means "the text so far fully matches".

class _RxAst:
    def __init__(self, kind, **kw):
        self.kind = kind
        self.__dict__.update(kw)


def _rx_parse(pattern: str) -> _RxAst:
    pos = 0

    def peek():
        return pattern[pos] if pos < len(pattern) else None

    def take():
        nonlocal pos
        ch = pattern[pos]
        pos += 1
        return ch

    def parse_alt():
        branches = [parse_concat()]
        while peek() == "|":
            take()
            branches.append(parse_concat())
        return branches[0] if len(branches) == 1 else _RxAst("alt", parts=branches)

    def parse_concat():
        parts = []
        while peek() is not None and peek() not in "|)":
            parts.append(parse_rep())
        if not parts:
            return _RxAst("empty")
        return parts[0] if len(parts) == 1 else _RxAst("cat", parts=parts)

    def parse_rep():
        atom = parse_atom()
        while True:
            c = peek()
            if c == "*":
                take()
                atom = _RxAst("star", inner=atom)
            elif c == "+":
                take()
                atom = _RxAst("cat", parts=[atom, _RxAst("star", inner=atom)])
            elif c == "?":
                take()
                atom = _RxAst("alt", parts=[atom, _RxAst("empty")])
            elif c == "{":
                take()
                spec = ""
                while peek() is not None and peek() != "}":
                    spec += take()
                if peek() != "}":
                    raise ValueError("unterminated {quantifier}")
                take()
                if "," in spec:
                    lo_s, hi_s = spec.split(",", 1)
                    lo = int(lo_s or 0)
                    hi = int(hi_s) if hi_s else None
                else:
                    lo = hi = int(spec)
                parts = [atom] * lo
                if hi is None:
                    parts.append(_RxAst("star", inner=atom))
                else:
                    parts.extend(_RxAst("alt", parts=[atom, _RxAst("empty")])
                                 for _ in range(hi - lo))
                atom = (_RxAst("empty") if not parts
                        else parts[0] if len(parts) == 1
                        else _RxAst("cat", parts=parts))
            else:
                return atom

    _CLASSES = {
        "d": lambda c: c.isdigit(),
        "D": lambda c: not c.isdigit(),
        "w": lambda c: c.isalnum() or c == "_",
        "W": lambda c: not (c.isalnum() or c == "_"),
        "s": lambda c: c in " \t\n\r\f\v",
        "S": lambda c: c not in " \t\n\r\f\v",
    }
    _ESCAPED = {"n": "\n", "t": "\t", "r": "\r"}

    def escape_pred():
        e = take()
        if e in _CLASSES:
            return _RxAst("pred", fn=_CLASSES[e], label=f"\\\\{e}")
        lit = _ESCAPED.get(e, e)
        return _RxAst("pred", fn=(lambda c, L=lit: c == L), label=lit)

    def parse_atom():
        c = take()
        if c == "(":
            inner = parse_alt()
            if peek() != ")":
                raise ValueError("unbalanced group")
            take()
            return inner
        if c == ".":
            return _RxAst("pred", fn=lambda ch: ch != "\n", label=".")
        if c == "[":
            neg = peek() == "^"
            if neg:
                take()
            items = []  # (lo, hi) ranges or predicate fns
            while peek() is not None and peek() != "]":
                a = take()
                if a == "\\\\" or a == "\\":
                    e = take()
                    if e in _CLASSES:
                        items.append(_CLASSES[e])
                        continue
                    a = _ESCAPED.get(e, e)
                if peek() == "-" and pos + 1 < len(pattern) \
                        and pattern[pos + 1] != "]":
                    take()
                    b = take()
                    items.append((a, b))
                else:
                    items.append((a, a))
            if peek() != "]":
                raise ValueError("unterminated character class")
            take()

            def in_class(ch, items=items, neg=neg):
                hit = any(it(ch) if callable(it) else it[0] <= ch <= it[1]
                          for it in items)
                return hit != neg

            return _RxAst("pred", fn=in_class, label="[class]")
        if c in ("\\", "\\\\"):
            return escape_pred()
        if c in "*+?{}|)":
            raise ValueError(f"unexpected {c!r}")
        return _RxAst("pred", fn=(lambda ch, L=c: ch == L), label=c)

    ast = parse_alt()
    if pos != len(pattern):
        raise ValueError(f"trailing regex input at {pos}")
    return ast


class RegexM:
    """NFA machine over a regex subset (advance/complete protocol)."""

    def __init__(self, pattern: str):
        # compile: states are ints; trans[i] = [(pred, j)]; eps[i] = [j]
        self.trans: list[list] = []
        self.eps: list[list[int]] = []

        def new_state():
            self.trans.append([])
            self.eps.append([])
            return len(self.trans) - 1

        def build(ast, start) -> int:
            """Wire ast from `start`; returns its accepting state."""
            if ast.kind == "empty":
                return start
            if ast.kind == "pred":
                end = new_state()
                self.trans[start].append((ast.fn, end))
                return end
            if ast.kind == "cat":
                cur = start
                for p in ast.parts:
                    cur = build(p, cur)
                return cur
            if ast.kind == "alt":
                end = new_state()
                for p in ast.parts:
                    s = new_state()
                    self.eps[start].append(s)
                    self.eps[build(p, s)].append(end)
                return end
            if ast.kind == "star":
                hub = new_state()
                self.eps[start].append(hub)
                s = new_state()
                self.eps[hub].append(s)
                self.eps[build(ast.inner, s)].append(hub)
                return hub
            raise AssertionError(ast.kind)

        s0 = new_state()
        self.accept = build(_rx_parse(pattern), s0)
        self.cur = self._closure({s0})

    def _closure(self, states: set) -> frozenset:
        stack, seen = list(states), set(states)
        while stack:
            for j in self.eps[stack.pop()]:
                if j not in seen:
                    seen.add(j)
                    stack.append(j)
        return frozenset(seen)

    @property
    def complete(self) -> bool:
        return self.accept in self.cur

    def advance(self, ch: str) -> bool:
        nxt = {j for i in self.cur for fn, j in self.trans[i] if fn(ch)}
        if not nxt:
            return False
        self.cur = self._closure(nxt)
        return True


def _regexm_clone(self) -> "RegexM":
    m = RegexM.__new__(RegexM)
    m.trans, m.eps, m.accept = self.trans, self.eps, self.accept  # static
    m.cur = self.cur
    return m


RegexM.clone = _regexm_clone


class GuidedRegexState:
    """Per-sequence guided_regex state (same probe/commit API as
    GuidedJsonState)."""

    def __init__(self, pattern: str):
        self.machine = RegexM(pattern)

    @property
    def complete(self) -> bool:
        return self.machine.complete

    def try_advance(self, text: str):
        m = self.machine.clone()
        for ch in text:
            if not m.advance(ch):
                return None
        return m

    def commit(self, machine) -> None:
        self.machine = machine


# ---------------------------------------------------------------------------
# guided_grammar: constrained decoding against a context-free grammar
# (reference surface: vLLM guided_grammar, Lark-style EBNF). First-party
# incremental EARLEY recognizer over characters: prefix-of-language
# membership is exactly "the next Earley state set is non-empty", which is
# the same probe/commit contract the JSON PDA and regex NFA use.
#
# Supported subset (covers the dialect's common shapes):
#   root: alt | alt            rule definitions with `:` or `::=`
#   "literal"                  quoted terminals (\\ \" \n \t escapes)
#   [a-z0-9_]                  character-class terminals (ranges + chars)
#   ( ... )  X* X+ X?          groups and postfix quantifiers
#   lowercase/UPPER rule refs; the start rule is `root` or `start` (or the
#   first rule defined)
# ---------------------------------------------------------------------------

class GrammarError(ValueError):
    pass


def _parse_grammar(text: str):
    """-> (rules: dict[name, list[list[sym]]], start). sym is
    ("r", name) | ("c", frozenset_of_chars)."""
    import re as _re

    rules: dict[str, list] = {}
    aux = [0]

    def fresh() -> str:
        aux[0] += 1
        return f"%aux{aux[0]}"

    def parse_class(body: str) -> frozenset:
        chars: set[str] = set()
        i = 0
        while i < len(body):
            if body[i] == "\\" and i + 1 < len(body):
                c = {"n": "\n", "t": "\t", "r": "\r"}.get(body[i + 1],
                                                          body[i + 1])
                chars.add(c)
                i += 2
            elif i + 2 < len(body) and body[i + 1] == "-":
                lo, hi = body[i], body[i + 2]
                chars.update(chr(x) for x in range(ord(lo), ord(hi) + 1))
                i += 3
            else:
                chars.add(body[i])
                i += 1
        return frozenset(chars)

    TOK = _re.compile(
        r'\s*(?:(?P<lit>"(?:\\.|[^"\\])*")'
        r'|(?P<cls>\[(?:\\.|[^\]\\])*\])'
        r'|(?P<ref>[A-Za-z_][A-Za-z0-9_]*)'
        r'|(?P<op>[()|*+?])'
        r')')

    def parse_alts(toks, pos, name):
        """toks: list of (kind, value); returns (alts, pos)."""
        alts, seq = [], []
        while pos < len(toks):
            kind, val = toks[pos]
            if kind == "op" and val == "|":
                alts.append(seq)
                seq = []
                pos += 1
                continue
            if kind == "op" and val == ")":
                break
            if kind == "op" and val == "(":
                sub, pos = parse_alts(toks, pos + 1, name)
                if pos >= len(toks) or toks[pos] != ("op", ")"):
                    raise GrammarError(f"unclosed group in rule {name!r}")
                pos += 1
                g = fresh()
                rules[g] = sub
                sym = ("r", g)
            elif kind == "lit":
                body = val[1:-1]
                out = []
                i = 0
                while i < len(body):
                    if body[i] == "\\" and i + 1 < len(body):
                        c = {"n": "\n", "t": "\t", "r": "\r"}.get(
                            body[i + 1], body[i + 1])
                        out.append(("c", frozenset((c,))))
                        i += 2
                    else:
                        out.append(("c", frozenset((body[i],))))
                        i += 1
                pos += 1
                # a multi-char literal is a sequence; quantifiers apply to
                # the whole literal via an aux rule
                if pos < len(toks) and toks[pos][0] == "op" \
                        and toks[pos][1] in "*+?":
                    g = fresh()
                    rules[g] = [out]
                    sym = ("r", g)
                else:
                    seq.extend(out)
                    continue
            elif kind == "cls":
                sym = ("c", parse_class(val[1:-1]))
                pos += 1
            elif kind == "ref":
                sym = ("r", val)
                pos += 1
            else:
                raise GrammarError(f"unexpected token {val!r} in {name!r}")
            # postfix quantifier
            if pos < len(toks) and toks[pos][0] == "op" \
                    and toks[pos][1] in "*+?":
                q = toks[pos][1]
                pos += 1
                g = fresh()
                if q == "*":
                    rules[g] = [[], [sym, ("r", g)]]
                elif q == "+":
                    rules[g] = [[sym], [sym, ("r", g)]]
                else:
                    rules[g] = [[], [sym]]
                sym = ("r", g)
            seq.append(sym)
        alts.append(seq)
        return alts, pos

    # split into rule definitions (a line continues until the next
    # `name:` at line start)
    lines = [ln for ln in text.splitlines()
             if ln.strip() and not ln.strip().startswith(("//", "#"))]
    defs: list[tuple[str, str]] = []
    head = _re.compile(r'^\s*([A-Za-z_][A-Za-z0-9_]*)\s*(?:::=|:)\s*(.*)$')
    for ln in lines:
        m = head.match(ln)
        if m and not defs or (m and not ln[:1].isspace()):
            defs.append((m.group(1), m.group(2)))
        elif defs:
            defs[-1] = (defs[-1][0], defs[-1][1] + " " + ln.strip())
        else:
            raise GrammarError(f"grammar must start with a rule: {ln!r}")
    if not defs:
        raise GrammarError("empty grammar")
    for name, body in defs:
        toks = []
        pos = 0
        while pos < len(body):
            m = TOK.match(body, pos)
            if not m:
                if body[pos:].strip() == "":
                    break
                raise GrammarError(
                    f"bad grammar syntax near {body[pos:pos + 20]!r}")
            pos = m.end()
            for kind in ("lit", "cls", "ref", "op"):
                if m.group(kind) is not None:
                    toks.append((kind, m.group(kind)))
                    break
        alts, end = parse_alts(toks, 0, name)
        if end != len(toks):
            raise GrammarError(f"unbalanced ')' in rule {name!r}")
        rules.setdefault(name, []).extend(alts)
    start = ("root" if "root" in rules
             else "start" if "start" in rules else defs[0][0])
    # validate refs
    for name, alts in rules.items():
        for alt in alts:
            for kind, val in alt:
                if kind == "r" and val not in rules:
                    raise GrammarError(f"undefined rule {val!r} "
                                       f"(referenced from {name!r})")
    return rules, start


class EarleyM:
    """Incremental Earley recognizer. Items are (rule, alt_idx, dot,
    origin); `sets[i]` is the closed state set after i characters. Older
    sets are immutable once built, so clone() is a shallow list copy —
    the probe/commit pattern stays cheap."""

    def __init__(self, rules, start):
        self.rules = rules
        self.start = start
        self.nullable = self._nullable(rules)
        s0 = self._closure({(start, a, 0, 0)
                            for a in range(len(rules[start]))}, 0, [])
        self.sets = [s0]

    @staticmethod
    def _nullable(rules) -> frozenset:
        """Rules that derive the empty string (fixed point)."""
        null = set()
        changed = True
        while changed:
            changed = False
            for name, alts in rules.items():
                if name in null:
                    continue
                for alt in alts:
                    if all(k == "r" and v in null for k, v in alt):
                        null.add(name)
                        changed = True
                        break
        return frozenset(null)

    def _closure(self, items: set, idx: int, sets) -> frozenset:
        rules = self.rules
        work = list(items)
        out = set(items)
        while work:
            rule, alt, dot, org = work.pop()
            body = rules[rule][alt]
            if dot < len(body):
                kind, val = body[dot]
                if kind == "r":  # predict
                    for a in range(len(rules[val])):
                        it = (val, a, 0, idx)
                        if it not in out:
                            out.add(it)
                            work.append(it)
                    if val in self.nullable:
                        # Aycock-Horspool: a nullable nonterminal may
                        # complete within this set regardless of item
                        # processing order — advance over it eagerly
                        it = (rule, alt, dot + 1, org)
                        if it not in out:
                            out.add(it)
                            work.append(it)
            else:  # complete: advance items waiting on `rule` at origin
                src = out if org == idx else sets[org]
                for r2, a2, d2, o2 in list(src):
                    b2 = rules[r2][a2]
                    if d2 < len(b2) and b2[d2] == ("r", rule):
                        it = (r2, a2, d2 + 1, o2)
                        if it not in out:
                            out.add(it)
                            work.append(it)
        return frozenset(out)

    def advance(self, ch: str) -> bool:
        idx = len(self.sets)
        scanned = set()
        for rule, alt, dot, org in self.sets[-1]:
            body = self.rules[rule][alt]
            if dot < len(body):
                kind, val = body[dot]
                if kind == "c" and ch in val:
                    scanned.add((rule, alt, dot + 1, org))
        if not scanned:
            return False
        self.sets.append(self._closure(scanned, idx, self.sets))
        return True

    @property
    def complete(self) -> bool:
        return any(r == self.start and o == 0
                   and d == len(self.rules[r][a])
                   for r, a, d, o in self.sets[-1])

    def clone(self) -> "EarleyM":
        m = EarleyM.__new__(EarleyM)
        m.rules, m.start, m.nullable = self.rules, self.start, self.nullable
        m.sets = self.sets[:]  # older sets are immutable
        return m


class GuidedGrammarState:
    """Per-sequence guided_grammar state (same probe/commit API as the
    JSON PDA and regex NFA states)."""

    def __init__(self, grammar: str):
        rules, start = _parse_grammar(grammar)
        self.machine = EarleyM(rules, start)

    @property
    def complete(self) -> bool:
        return self.machine.complete

    def try_advance(self, text: str):
        m = self.machine.clone()
        for ch in text:
            if not m.advance(ch):
                return None
        return m

    def commit(self, machine) -> None:
        self.machine = machine
