"""Guided JSON decoding: char-level PDA, schema templates, engine + server
integration (reference: vLLM guided_json through gpustack backend params)."""
import json

import pytest

from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
from gpustack_amd.engine.guided import (
    GuidedJsonState, JsonPDA, compile_schema,
)


def _accepts(machine, text: str) -> bool:
    for ch in text:
        if not machine.advance(ch):
            return False
    return True


@pytest.mark.parametrize("text", [
    '{"a": 1}',
    '[1, 2.5, {"x": [true, null]}]',
    '"hi\\n there \\u00e9"',
    '-0.5e3',
    'true',
    '{}',
    '[]',
    '{"nested": {"deep": [[]]}}',
    '  {"ws" : [ 1 , 2 ] }',
])
def test_pda_accepts_valid(text):
    m = JsonPDA()
    assert _accepts(m, text)
    assert m.complete


@pytest.mark.parametrize("text", [
    '{,',
    '[1,]',
    '{"a" 1}',
    '01',
    'tru x',
    '{"a":}',
    '}',
    '{"a": 1,}',
])
def test_pda_rejects_invalid(text):
    assert not _accepts(JsonPDA(), text)


def test_pda_incomplete_prefixes():
    m = JsonPDA()
    assert _accepts(m, '{"key": [1, 2')
    assert not m.complete
    assert _accepts(m, ']}')
    assert m.complete


SCHEMA = {
    "type": "object",
    "properties": {
        "name": {"type": "string"},
        "age": {"type": "integer"},
        "ok": {"type": "boolean"},
        "tags": {"type": "array", "items": {"type": "string"}},
        "mode": {"enum": ["fast", "slow"]},
    },
}


def test_schema_template_accepts_canonical():
    m = compile_schema(SCHEMA)
    good = '{"name":"bo","age":42,"ok":true,"tags":["a","b"],"mode":"slow"}'
    assert _accepts(m, good)
    assert m.complete
    assert json.loads(good)


@pytest.mark.parametrize("bad", [
    '{"age":42',                  # wrong first key
    '{"name":42',                 # wrong type
    '{"name":"bo","age":4.5',     # float for integer
    '{"name":"bo","age":42,"ok":"y"',   # string for boolean
    '{"name":"bo","age":42,"ok":true,"tags":[1',  # non-string item
    '{"name":"bo","age":42,"ok":true,"tags":[],"mode":"mid"',  # bad enum
])
def test_schema_template_rejects(bad):
    assert not _accepts(compile_schema(SCHEMA), bad)


def _engine_with_table(**kw):
    from gpustack_amd.worker.engine_server import ByteTokenizer

    kw.setdefault("model", "tiny")
    kw.setdefault("device", "cpu")
    kw.setdefault("kv_cache_blocks", 64)
    eng = LLMEngine(EngineConfig(**kw))
    tok = ByteTokenizer(eng.cfg.spec.vocab_size)
    eng.set_token_table([tok.decode([i]) for i in range(eng.cfg.spec.vocab_size)])
    return eng, tok


def test_engine_guided_any_json_prefix_valid():
    """Every emitted token keeps the output a valid JSON prefix; if the
    request finished by stop, the whole output parses."""
    eng, tok = _engine_with_table()
    p = SamplingParams(max_tokens=48, guided_json=True, eos_token_id=1)
    rid = eng.add_request([20, 21, 22], p)
    toks, reason = [], None
    while eng.has_unfinished():
        for o in eng.step():
            if o.request_id == rid:
                if o.finish_reason:
                    reason = o.finish_reason
                toks.append(o.token_id)
    text = tok.decode([t for t in toks if t != 1])
    m = JsonPDA()
    assert _accepts(m, text), f"invalid prefix: {text!r}"
    if reason == "stop":
        json.loads(text)


def test_engine_guided_schema_exact():
    eng, tok = _engine_with_table()
    schema = {"type": "object",
              "properties": {"ok": {"type": "boolean"},
                             "mode": {"enum": ["a", "b"]}}}
    p = SamplingParams(max_tokens=40, guided_json=schema, eos_token_id=1)
    out = eng.generate([[30, 31]], p)[0]
    text = tok.decode([t for t in out if t != 1])
    doc = json.loads(text)
    assert set(doc) == {"ok", "mode"}
    assert isinstance(doc["ok"], bool)
    assert doc["mode"] in ("a", "b")


def test_engine_guided_sampled_schema():
    """Temperature sampling stays inside the grammar too."""
    eng, tok = _engine_with_table()
    schema = {"type": "object", "properties": {"ok": {"type": "boolean"}}}
    p = SamplingParams(max_tokens=30, temperature=0.9, seed=3,
                       guided_json=schema, eos_token_id=1)
    out = eng.generate([[40]], p)[0]
    doc = json.loads(tok.decode([t for t in out if t != 1]))
    assert isinstance(doc["ok"], bool)


@pytest.mark.timeout(240)
def test_engine_server_response_format():
    import socket
    import subprocess
    import sys
    import time

    import httpx

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-g", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    try:
        t0 = time.time()
        while time.time() - t0 < 90:
            if proc.poll() is not None:
                raise AssertionError(f"engine server exited {proc.returncode}")
            try:
                if httpx.get(f"http://127.0.0.1:{port}/health",
                             timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                pass
            time.sleep(0.5)
        else:
            raise AssertionError("engine server never became healthy")
        r = httpx.post(f"http://127.0.0.1:{port}/v1/completions", json={
            "model": "tiny-g", "prompt": "give me json", "max_tokens": 40,
            "temperature": 0,
            "response_format": {"type": "json_schema", "json_schema": {
                "schema": {"type": "object",
                           "properties": {"ok": {"type": "boolean"}}}}},
        }, timeout=120)
        assert r.status_code == 200
        doc = json.loads(r.json()["choices"][0]["text"])
        assert isinstance(doc["ok"], bool)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


@pytest.mark.timeout(240)
def test_tool_calling():
    """OpenAI tool calling: forced tool_choice constrains arguments with the
    function's JSON-Schema; "required" uses a name+arguments envelope."""
    import socket
    import subprocess
    import sys
    import time

    import httpx

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    proc = subprocess.Popen([
        sys.executable, "-m", "gpustack_amd.worker.engine_server",
        "--served-name", "tiny-t", "--source", "preset", "--model-ref", "tiny",
        "--port", str(port), "--max-model-len", "256",
        "--device", "cpu", "--kv-cache-blocks", "64",
    ])
    base = f"http://127.0.0.1:{port}"
    # maxLength bounds keep the random-init model's strings finite
    tools = [{"type": "function", "function": {
        "name": "get_weather",
        "parameters": {"type": "object",
                       "properties": {"city": {"type": "string",
                                               "maxLength": 12},
                                      "celsius": {"type": "boolean"}}}}},
        {"type": "function", "function": {
            "name": "get_time",
            "parameters": {"type": "object",
                           "properties": {"tz": {"type": "string",
                                                 "maxLength": 12}}}}}]
    try:
        t0 = time.time()
        while time.time() - t0 < 90:
            if proc.poll() is not None:
                raise AssertionError(f"engine server exited {proc.returncode}")
            try:
                if httpx.get(f"{base}/health", timeout=2).status_code == 200:
                    break
            except httpx.HTTPError:
                pass
            time.sleep(0.5)
        else:
            raise AssertionError("engine server never became healthy")

        msg = [{"role": "user", "content": "weather in Oslo?"}]
        # forced function
        r = httpx.post(f"{base}/v1/chat/completions", json={
            "model": "tiny-t", "messages": msg, "max_tokens": 60,
            "temperature": 0, "tools": tools,
            "tool_choice": {"type": "function",
                            "function": {"name": "get_weather"}},
        }, timeout=120)
        assert r.status_code == 200, r.text
        ch = r.json()["choices"][0]
        assert ch["finish_reason"] == "tool_calls"
        tc = ch["message"]["tool_calls"][0]
        assert tc["function"]["name"] == "get_weather"
        args = json.loads(tc["function"]["arguments"])
        assert set(args) == {"city", "celsius"}
        assert isinstance(args["city"], str) and isinstance(args["celsius"], bool)

        # required: envelope picks one of the declared tools
        r = httpx.post(f"{base}/v1/chat/completions", json={
            "model": "tiny-t", "messages": msg, "max_tokens": 80,
            "temperature": 0, "tools": tools, "tool_choice": "required",
        }, timeout=120)
        ch = r.json()["choices"][0]
        assert ch["finish_reason"] == "tool_calls"
        tc = ch["message"]["tool_calls"][0]
        assert tc["function"]["name"] in ("get_weather", "get_time")
        json.loads(tc["function"]["arguments"])

        # tool_choice none: plain content answer
        r = httpx.post(f"{base}/v1/chat/completions", json={
            "model": "tiny-t", "messages": msg, "max_tokens": 8,
            "temperature": 0, "ignore_eos": True,
            "tools": tools, "tool_choice": "none",
        }, timeout=120)
        ch = r.json()["choices"][0]
        assert ch["message"].get("content")
        assert "tool_calls" not in ch["message"]

        # unknown forced tool -> 400
        r = httpx.post(f"{base}/v1/chat/completions", json={
            "model": "tiny-t", "messages": msg, "tools": tools,
            "tool_choice": {"type": "function", "function": {"name": "nope"}},
        }, timeout=60)
        assert r.status_code == 400

        # streaming: single tool_calls delta then finish
        deltas = []
        with httpx.stream("POST", f"{base}/v1/chat/completions", json={
            "model": "tiny-t", "messages": msg, "max_tokens": 60,
            "temperature": 0, "stream": True, "tools": tools,
            "tool_choice": {"type": "function",
                            "function": {"name": "get_weather"}},
        }, timeout=120) as resp:
            for line in resp.iter_lines():
                if line.startswith("data:") and "[DONE]" not in line:
                    deltas.append(json.loads(line[5:]))
        tc_deltas = [d for d in deltas
                     if d["choices"][0]["delta"].get("tool_calls")]
        assert len(tc_deltas) == 1
        args = json.loads(tc_deltas[0]["choices"][0]["delta"]["tool_calls"][0]
                          ["function"]["arguments"])
        assert set(args) == {"city", "celsius"}
        assert deltas[-1]["choices"][0]["finish_reason"] == "tool_calls"
        # no plain-content deltas leaked in tool mode
        assert not any(d["choices"][0]["delta"].get("content") for d in deltas)
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()


def test_array_of_overlapping_enums():
    """Ambiguous enum items ("a" vs "ab") keep extending inside arrays."""
    schema = {"type": "object",
              "properties": {"xs": {"type": "array",
                                    "items": {"enum": ["a", "ab"]}}}}
    assert _accepts(compile_schema(schema), '{"xs":["ab","a","ab"]}')
    assert _accepts(compile_schema(schema), '{"xs":[]}')
    assert not _accepts(compile_schema(schema), '{"xs":["b"]}')
    m = compile_schema(schema)
    assert _accepts(m, '{"xs":["a"]}') and m.complete


import pytest as _pytest


@_pytest.mark.parametrize("pattern,good,bad", [
    (r"[a-c]+x", "abcx", "abd"),
    (r"(ab|cd){2}", "abcd", "abx"),
    (r"\d{3}-\d{4}", "555-1234", "55-1234x"),
    (r"yes|no|maybe", "maybe", "nope?"),
    (r"a?b+c", "bbc", "ac?"),
    (r"[^0-9]+", "hello", "h3"),
    (r"\w+@\w+\.(com|org)", "a_1@b.org", "a@b.net?"),
])
def test_regex_machine(pattern, good, bad):
    from gpustack_amd.engine.guided import RegexM

    def feed(s):
        m = RegexM(pattern)
        for ch in s:
            if not m.advance(ch):
                return None
        return m

    m = feed(good)
    assert m is not None and m.complete
    assert feed(bad) is None or not feed(bad).complete


def test_regex_prefix_validity():
    from gpustack_amd.engine.guided import RegexM

    m = RegexM(r"\d{3}-\d{4}")
    for ch in "555-":
        assert m.advance(ch)
    assert not m.complete      # valid prefix, not a full match
    assert not m.advance("x")  # invalid continuation rejected


def test_engine_guided_regex():
    eng, tok = _engine_with_table()
    p = SamplingParams(max_tokens=20, guided_regex=r"(red|green|blue)!",
                       eos_token_id=1)
    out = eng.generate([[25, 26]], p)[0]
    text = tok.decode([t for t in out if t != 1])
    assert text in ("red!", "green!", "blue!")


def test_guided_regex_not_shortest_match():
    """when the machine is complete but still extensible, EOS
    competes on logits with continuation tokens instead of being forced
    ('\\d+' must be able to emit more than one digit)."""
    import torch

    from gpustack_amd.engine.model_runner import Sampler
    from gpustack_amd.engine.sequence import SamplingParams, Sequence

    table = ["<eos>", "1", "2", "a"]
    s = Sampler("cpu")
    s.token_table = table
    p = SamplingParams(guided_regex=r"\d+", eos_token_id=0)

    # digit logit above EOS: generation continues past the first digit
    seq = Sequence("t", [9], p)
    seq.output_token_ids = ["skip"]  # placeholder replaced below
    seq.output_token_ids = [1]       # one digit emitted -> machine complete
    row = torch.tensor([1.0, 5.0, 0.0, 9.0])  # 'a' highest but invalid
    assert s._guided_pick(row, seq) == 1      # continues with '1'

    # EOS logit above all digits: terminates (complete => EOS valid)
    seq2 = Sequence("t2", [9], p)
    seq2.output_token_ids = [1]
    row2 = torch.tensor([5.0, 1.0, 0.0, 9.0])
    assert s._guided_pick(row2, seq2) == 0

    # machine NOT complete (no digit yet): EOS is not a valid pick
    seq3 = Sequence("t3", [9], p)
    row3 = torch.tensor([9.0, 1.0, 0.0, 5.0])
    assert s._guided_pick(row3, seq3) == 1


# ---- guided_grammar (EBNF CFG via incremental Earley, engine/guided.py) ----

def test_grammar_parse_and_membership():
    from gpustack_amd.engine.guided import GrammarError, GuidedGrammarState

    g = GuidedGrammarState('''
// arithmetic over ints
root: expr
expr: term (("+" | "-") term)*
term: [0-9]+ | "(" expr ")"
''')
    assert g.try_advance("(1+2)-30") is not None
    assert g.try_advance("(1+2)-30").complete
    assert g.try_advance("(1+2") is not None          # extensible prefix
    assert not g.try_advance("(1+2").complete
    assert g.try_advance("1++") is None               # dead prefix
    assert g.try_advance("x") is None
    import pytest as _pytest

    with _pytest.raises(GrammarError):
        GuidedGrammarState("root: undefined_rule")
    with _pytest.raises(GrammarError):
        GuidedGrammarState("")


def test_grammar_quantifiers_and_literals():
    from gpustack_amd.engine.guided import GuidedGrammarState

    g = GuidedGrammarState('''
root: "ab"+ tail?
tail: ";" [xyz]*
''')
    assert g.try_advance("abab;xy").complete
    assert g.try_advance("ab").complete               # tail optional
    assert g.try_advance("aba") is not None           # mid-literal
    assert g.try_advance("ba") is None
    assert g.try_advance("ab;").complete              # [xyz]* empty


def test_grammar_nested_recursion():
    from gpustack_amd.engine.guided import GuidedGrammarState

    g = GuidedGrammarState('root: "(" root ")" | ""')
    assert g.try_advance("((()))").complete
    assert g.try_advance("((").complete is False
    assert g.try_advance("((") is not None
    assert g.try_advance(")") is None


def test_guided_grammar_generation_end_to_end():
    """Engine generation constrained by a CFG: every emitted string is a
    valid prefix and the final output parses completely."""
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.engine.guided import GuidedGrammarState

    eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64, max_model_len=128,
                                 seed=0))
    # token table: single printable chars so grammar chars are reachable
    table = [""] * eng.cfg.spec.vocab_size
    for i in range(32, 127):
        table[i] = chr(i)
    eng.set_token_table(table)
    # bounded language: guided decoding keeps every prefix valid but (by
    # design, non-shortest-match) only finishes when the grammar forces
    # it or EOS wins on logits — so the completion check uses a grammar
    # whose strings have bounded length
    grammar = '''
root: "{" pair "}"
pair: [a-z] [a-z]? "=" [0-9] [0-9]?
'''
    p = SamplingParams(max_tokens=16, guided_grammar=grammar,
                       eos_token_id=1, temperature=0.0)
    toks = eng.generate([[5, 9, 2]], p)[0]
    text = "".join(table[t] for t in toks if t != 1)
    m = GuidedGrammarState(grammar).try_advance(text)
    assert m is not None, text
    assert m.complete, f"incomplete guided output: {text!r}"


def test_guided_grammar_sampled_generation():
    from gpustack_amd.engine import EngineConfig, LLMEngine, SamplingParams
    from gpustack_amd.engine.guided import GuidedGrammarState

    eng = LLMEngine(EngineConfig(model="tiny", device="cpu",
                                 kv_cache_blocks=64, max_model_len=128,
                                 seed=0))
    table = [""] * eng.cfg.spec.vocab_size
    for i in range(32, 127):
        table[i] = chr(i)
    eng.set_token_table(table)
    grammar = 'root: ("yes" | "no" | "maybe")'
    p = SamplingParams(max_tokens=8, guided_grammar=grammar,
                       eos_token_id=1, temperature=0.9, seed=7)
    toks = eng.generate([[4, 4]], p)[0]
    text = "".join(table[t] for t in toks if t != 1)
    assert text in ("yes", "no", "maybe"), text


def test_grammar_nullable_chains():
    """Aycock-Horspool nullable handling: nonterminals deriving empty must
    complete regardless of item processing order."""
    from gpustack_amd.engine.guided import GuidedGrammarState

    g = GuidedGrammarState('''
root: a "x"
a: b b
b: ""
''')
    assert g.try_advance("x").complete
    g2 = GuidedGrammarState('root: a a "y"\na: "" | "w"')
    assert g2.try_advance("y").complete
    assert g2.try_advance("wy").complete
    assert g2.try_advance("wwy").complete
    assert g2.try_advance("wwwy") is None
