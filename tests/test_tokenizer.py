"""ByteTokenizer: round-trips, chat-template structure, tool rendering."""
import pytest

from agentcontrolplane_amd.engine.tokenizer import (
    BOS, END_HEADER, EOT, N_SPECIAL, START_HEADER, ByteTokenizer,
)


@pytest.fixture
def tok():
    return ByteTokenizer(512)


def test_text_roundtrip_exact(tok):
    for s in ["", "hello", "héllo wörld", "日本語 🙂", "a\nb\tc", '"quotes\\"']:
        assert tok.decode(tok.encode_text(s)) == s


def test_decode_skips_specials(tok):
    ids = tok.encode_text("ab") + [EOT, BOS] + tok.encode_text("cd")
    assert tok.decode(ids) == "abcd"


def test_chat_template_structure(tok):
    ids = tok.render_chat(
        [{"role": "system", "content": "sys"}, {"role": "user", "content": "usr"}]
    )
    assert ids[0] == BOS
    # two headers, two EOTs, then the assistant header is opened for generation
    assert ids.count(START_HEADER) >= 3 and ids.count(END_HEADER) >= 3
    assert ids.count(EOT) == 2
    assert all(0 <= t < N_SPECIAL for t in ids)
    # role bytes appear between header markers
    text = tok.decode(ids)
    assert "system" in text and "user" in text and "sys" in text and "usr" in text


def test_tools_rendered_into_system(tok):
    tools = [{"type": "function", "function": {
        "name": "srv__add", "description": "add numbers",
        "parameters": {"type": "object", "properties": {"a": {"type": "number"}}}}}]
    with_tools = tok.render_chat([{"role": "user", "content": "x"}], tools)
    without = tok.render_chat([{"role": "user", "content": "x"}])
    assert len(with_tools) > len(without)
    assert "srv__add" in tok.decode(with_tools)


def test_tool_result_and_calls_render(tok):
    msgs = [
        {"role": "user", "content": "go"},
        {"role": "assistant", "toolCalls": [{"id": "c1", "type": "function",
         "function": {"name": "t__x", "arguments": '{"k": 1}'}}]},
        {"role": "tool", "content": "42", "toolCallId": "c1"},
    ]
    ids = tok.render_chat(msgs)
    text = tok.decode(ids)
    assert "t__x" in text and "42" in text


def test_vocab_floor():
    with pytest.raises(ValueError):
        ByteTokenizer(100)
