"""Constrained-decoding grammar: every random walk through the automaton must
produce parseable tool-call JSON naming an offered tool."""
import json
import random

import pytest

from agentcontrolplane_amd.engine.grammar import JsonValueMachine, ToolCallGrammar
from agentcontrolplane_amd.engine.tokenizer import EOT


def drive(grammar, rng, max_steps=3000):
    steps = 0
    while not grammar.finished and steps < max_steps:
        allowed = grammar.allowed_tokens()
        assert allowed, f"dead end in phase {grammar.phase}"
        tok = rng.choice(sorted(allowed))
        grammar.advance(tok)
        steps += 1
        if grammar.phase == "done":
            grammar.advance(EOT)
    assert grammar.finished
    return bytes(grammar.buf).decode()


@pytest.mark.parametrize("seed", range(20))
def test_random_walks_parse(seed):
    rng = random.Random(seed)
    g = ToolCallGrammar(["calc__add", "calc__echo", "search__web"], max_args_len=200)
    text = drive(g, rng)
    obj = json.loads(text)
    assert obj["name"] in ("calc__add", "calc__echo", "search__web")
    assert isinstance(obj["arguments"], dict)
    name, args = g.parse()
    assert name == obj["name"]
    json.loads(args)


def test_forced_sequence():
    g = ToolCallGrammar(["add"])
    target = b'{"name": "add", "arguments": {"a": 1, "b": 2.5}}'
    for b in target:
        assert b in g.allowed_tokens(), f"byte {chr(b)!r} rejected, phase={g.phase}"
        g.advance(b)
    assert g.accepting
    assert EOT in g.allowed_tokens()
    g.advance(EOT)
    name, args = g.parse()
    assert name == "add"
    assert json.loads(args) == {"a": 1, "b": 2.5}


def test_shared_prefix_tool_names():
    g = ToolCallGrammar(["calc__add", "calc__add_more"])
    for b in b'{"name": "calc__add':
        g.advance(b)
    # both continuing and closing must be legal at the shared prefix
    allowed = g.allowed_tokens()
    assert ord('"') in allowed and ord("_") in allowed


def test_illegal_name_rejected():
    g = ToolCallGrammar(["add"])
    for b in b'{"name": "':
        g.advance(b)
    assert ord("z") not in g.allowed_tokens()


def test_json_value_machine_nested():
    m = JsonValueMachine(root_object=True)
    for b in b'{"x": [1, {"y": "z\\n"}, true, null, -2.5e3]}':
        assert b in m.allowed(), f"rejected {chr(b)!r} in state {m.state}"
        m.advance(b)
    assert m.done


def test_json_depth_limit():
    m = JsonValueMachine(max_depth=2, root_object=True)
    for b in b'{"a": {':
        m.advance(b)
    # at depth 2, '{' and '[' are no longer offered for values
    allowed = m.allowed()
    for b in b'"k':
        pass
    m.advance(ord('"'))
    for b in b'k": ':
        m.advance(b)
    assert ord("{") not in m.allowed() and ord("[") not in m.allowed()


@pytest.mark.parametrize("seed", range(10))
def test_schema_constrained_arguments(seed):
    """Schema-aware args: required keys, correct types — executable calls."""
    rng = random.Random(seed)
    tools = [
        {"type": "function", "function": {
            "name": "delegate_to_agent__worker",
            "parameters": {"type": "object",
                           "properties": {"message": {"type": "string"}},
                           "required": ["message"]}}},
        {"type": "function", "function": {
            "name": "calc__add",
            "parameters": {"type": "object",
                           "properties": {"a": {"type": "number"}, "b": {"type": "number"}},
                           "required": ["a", "b"]}}},
        {"type": "function", "function": {
            "name": "flags__set",
            "parameters": {"type": "object",
                           "properties": {"on": {"type": "boolean"}},
                           "required": ["on"]}}},
    ]
    g = ToolCallGrammar(tools=tools, max_args_len=80)
    text = drive(g, rng)
    obj = json.loads(text)
    name, args = obj["name"], obj["arguments"]
    if name == "delegate_to_agent__worker":
        assert isinstance(args["message"], str)
        assert set(args) == {"message"}
    elif name == "calc__add":
        assert isinstance(args["a"], (int, float)) and isinstance(args["b"], (int, float))
        assert set(args) == {"a", "b"}
    else:
        assert isinstance(args["on"], bool)


def test_schema_free_form_fallback():
    """A tool without properties keeps the free-form object grammar."""
    tools = [{"type": "function", "function": {"name": "t__x", "parameters": {}}}]
    rng = random.Random(0)
    g = ToolCallGrammar(tools=tools, max_args_len=60)
    obj = json.loads(drive(g, rng))
    assert isinstance(obj["arguments"], dict)


def test_schema_keys_and_tool_names_with_json_specials():
    """Property names and tool names containing quotes/backslashes must be
    JSON-escaped in the forced output (raw emission produced invalid JSON)."""
    import json
    import random

    from agentcontrolplane_amd.engine.tokenizer import EOT

    cases = [
        ({'a"b': {"type": "string"}}, "t__x"),
        ({"a\\b": {"type": "integer"}}, "t__x"),
        ({"ok": {"type": "boolean"}}, 'we"ird\\name'),
    ]
    for props, name in cases:
        tools = [{"type": "function", "function": {
            "name": name,
            "parameters": {"type": "object", "properties": props,
                           "required": list(props)}}}]
        g = ToolCallGrammar(tools=tools, max_args_len=64)
        rng = random.Random(3)
        for _ in range(4000):
            if g.accepting:
                g.advance(EOT)
                break
            g.advance(rng.choice(sorted(g.allowed_tokens())))
        n, args = g.parse()
        assert n == name
        parsed = json.loads(args)
        assert set(parsed) == set(props)


def test_utf8_lead_near_budget_always_closes_valid():
    """Regression (r02 GPU bench): budget exhaustion right after a UTF-8
    lead byte must FINISH the codepoint on the closing path — an empty
    allowed set here let the sampler emit byte 0 mid-sequence and
    parse() crashed on invalid UTF-8."""
    import random

    # tight budgets exercise the closing path; roomy budgets exercise
    # long string values where lead->continuation bookkeeping must hold
    budgets = [lambda r: r.randint(8, 24)] * 300 + [lambda r: 100] * 100
    for seed, budget in enumerate(budgets):
        rng = random.Random(seed)
        g = ToolCallGrammar(tool_names=["t"], max_args_len=budget(rng))
        steps = 0
        while not g.finished and steps < 600:
            allowed = g.allowed_tokens()
            assert allowed, f"empty allowed set in phase {g.phase} (seed {seed})"
            # bias toward UTF-8 leads so budget exhaustion lands mid-codepoint
            leads = [b for b in allowed if 0xC2 <= b < 0xF5]
            tok = rng.choice(leads) if leads and rng.random() < 0.6 else rng.choice(
                sorted(allowed)
            )
            g.advance(tok)
            steps += 1
            if g.phase == "done":
                g.advance(EOT)
        assert g.finished, f"seed {seed} never finished"
        name, args = g.parse()  # must decode + json-parse cleanly
        assert name == "t"
