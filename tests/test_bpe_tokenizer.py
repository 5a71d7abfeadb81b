"""BPE tokenizer + token-level constrained decoding (token-trie masks).

A small byte-level BPE tokenizer is trained in-process (the `tokenizers`
package; no network) and saved as tokenizer.json — the same artifact shape
a real Llama-3 checkpoint directory carries, exercising the exact loading
path (EngineConfig.tokenizer_path / <checkpoint>/tokenizer.json)."""
from __future__ import annotations

import json

import pytest

from agentcontrolplane_amd.engine.config import EngineConfig
from agentcontrolplane_amd.engine.engine import InferenceEngine
from agentcontrolplane_amd.engine.grammar import ToolCallGrammar
from agentcontrolplane_amd.engine.request import SamplingParams
from agentcontrolplane_amd.engine.token_grammar import TokenGrammar, TokenTrie
from agentcontrolplane_amd.engine.tokenizer import HFTokenizer


@pytest.fixture(scope="module")
def bpe_path(tmp_path_factory):
    from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=420, special_tokens=[], show_progress=False,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    corpus = [
        '{"name": "calc__add", "arguments": {"a": 1, "b": 2}}',
        '{"name": "tools__echo", "arguments": {"text": "héllo wörld"}}',
        "the quick brown fox jumps over the lazy dog",
        "assistant system user tool message content",
    ] * 50
    tok.train_from_iterator(corpus, trainer)
    path = tmp_path_factory.mktemp("bpe") / "tokenizer.json"
    tok.save(str(path))
    return str(path)


def test_hf_tokenizer_roundtrip(bpe_path):
    t = HFTokenizer(bpe_path)
    for text in ("hello world", '{"a": 1}', "héllo wörld ünïcode", "日本"):
        ids = t.encode_text(text)
        assert t.decode(ids) == text
    # specials were added and are distinct
    assert len({t.bos, t.start_header, t.end_header, t.eot}) == 4
    assert t.live_vocab >= 260  # byte alphabet + merges + specials
    # multi-byte tokens exist (BPE merged something)
    assert any(
        t.token_bytes(i) is not None and len(t.token_bytes(i)) > 1
        for i in range(t.live_vocab)
    )
    # chat render uses the specials and round-trips the content
    ids = t.render_chat(
        [{"role": "user", "content": "hi"}],
        tools=[{"type": "function", "function": {"name": "f"}}],
    )
    assert ids[0] == t.bos and t.start_header in ids


def test_token_trie_walk_matches_bruteforce(bpe_path):
    """The trie walk's mask == brute force over every token's bytes."""
    t = HFTokenizer(bpe_path)
    tools = [{"type": "function", "function": {
        "name": "calc__add",
        "parameters": {"type": "object",
                       "properties": {"a": {"type": "number"}},
                       "required": ["a"]}}}]
    tb = [t.token_bytes(i) for i in range(t.live_vocab)]
    trie = TokenTrie(tb)

    def brute(byte_grammar):
        out = set()
        for tid, bs in enumerate(tb):
            if not bs:
                continue
            g = byte_grammar.clone()
            ok = True
            for b in bs:
                if b not in g.allowed_tokens():
                    ok = False
                    break
                g.advance(b)
            if ok:
                out.add(tid)
        return out

    tg = TokenGrammar(trie, tb, t.eot, tools, max_args_len=256)
    # check at several points along a generation
    for _ in range(12):
        mask = tg.allowed_tokens()
        assert mask == brute(tg.byte_grammar)
        assert mask, f"dead end in phase {tg.phase}"
        # advance with the smallest allowed token (deterministic)
        tg.advance(min(mask))
        if tg.phase == "done":
            break


def test_bpe_constrained_tool_call_e2e(bpe_path):
    """CPU engine + BPE tokenizer: a constrained turn emits a grammar-valid
    executable tool call through token-level masks (multi-byte tokens)."""
    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=512,
                     tokenizer_path=bpe_path, max_prefill_tokens=512),
        start=True,
    )
    try:
        assert eng.scheduler.grammar_factory is not None  # token-trie active
        tools = [{"type": "function", "function": {
            "name": "calc__add",
            "description": "add numbers",
            "parameters": {"type": "object",
                           "properties": {"a": {"type": "number"},
                                          "b": {"type": "number"}},
                           "required": ["a", "b"]}}}]
        res = eng.chat(
            [{"role": "user", "content": "add 1 and 2"}],
            tools=tools,
            sampling=SamplingParams(max_tokens=64, temperature=0.9,
                                    tool_choice="required"),
        )
        assert res.finish_reason == "tool_calls", res.finish_reason
        call = res.tool_calls[0]["function"]
        assert call["name"] == "calc__add"
        args = json.loads(call["arguments"])
        assert set(args) == {"a", "b"}
        assert isinstance(args["a"], (int, float))
    finally:
        eng.stop()


def test_bpe_free_text_turn(bpe_path):
    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=512,
                     tokenizer_path=bpe_path, max_prefill_tokens=512),
        start=True,
    )
    try:
        res = eng.chat(
            [{"role": "user", "content": "say something"}],
            sampling=SamplingParams(max_tokens=12, temperature=1.0),
        )
        assert res.completion_tokens <= 12
        assert isinstance(res.text, str)
    finally:
        eng.stop()


def test_utf8_constrained_string(bpe_path):
    """Multi-byte UTF-8 inside constrained string arguments decodes to
    valid text (VERDICT weak #5: STRING_SAFE previously forbade it)."""
    t = HFTokenizer(bpe_path)
    tools = [{"type": "function", "function": {
        "name": "tools__echo",
        "parameters": {"type": "object",
                       "properties": {"text": {"type": "string"}},
                       "required": ["text"]}}}]
    tb = [t.token_bytes(i) for i in range(t.live_vocab)]
    trie = TokenTrie(tb)
    tg = TokenGrammar(trie, tb, t.eot, tools, max_args_len=128)
    # drive to the string-value position
    import random

    rng = random.Random(7)
    # walk until inside the argument string, then force UTF-8 content
    utf8_token = next(
        i for i in range(t.live_vocab)
        if tb[i] is not None and any(b >= 0xC2 for b in tb[i])
        and t.decode([i]) != "�"
    )
    injected = False
    for _ in range(200):
        if tg.finished:
            break
        if tg.accepting:
            # the engine adds EOT at accepting states (allowed_tokens only
            # returns byte-backed tokens)
            tg.advance(t.eot)
            break
        mask = tg.allowed_tokens()
        assert mask, f"dead end in phase {tg.phase}"
        if not injected and utf8_token in mask:
            tg.advance(utf8_token)
            injected = True
            continue
        tg.advance(rng.choice(sorted(mask)))
    assert tg.finished
    name, args = tg.parse()
    assert name == "tools__echo"
    text = json.loads(args)["text"]
    assert isinstance(text, str)  # parse() already utf-8 decoded the buffer
    if injected:
        assert "�" not in text  # no mojibake: UTF-8 stayed well-formed


def test_checkpoint_dir_with_tokenizer_json(bpe_path, tmp_path):
    """Full serving path for a real checkpoint directory: safetensors
    weights + tokenizer.json side by side — the engine loads both and
    constrained decoding runs token-level (models/weights.py +
    EngineConfig.checkpoint_path auto-detecting tokenizer.json)."""
    import shutil

    from agentcontrolplane_amd.models import create_model
    from agentcontrolplane_amd.engine.config import PRESETS
    from agentcontrolplane_amd.models.weights import save_checkpoint

    ckpt = tmp_path / "ckpt"
    model = create_model(PRESETS["tiny"], EngineConfig(model="tiny", device="cpu"), "cpu")
    model.random_init(99)
    save_checkpoint(model, str(ckpt))
    shutil.copy(bpe_path, ckpt / "tokenizer.json")

    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=512,
                     checkpoint_path=str(ckpt), max_prefill_tokens=512),
        start=True,
    )
    try:
        assert isinstance(eng.tokenizer, HFTokenizer)
        tools = [{"type": "function", "function": {
            "name": "tools__echo",
            "parameters": {"type": "object",
                           "properties": {"text": {"type": "string"}},
                           "required": ["text"]}}}]
        res = eng.chat(
            [{"role": "user", "content": "echo hi"}],
            tools=tools,
            sampling=SamplingParams(max_tokens=48, temperature=0.8,
                                    tool_choice="required"),
        )
        assert res.finish_reason == "tool_calls"
        assert res.tool_calls[0]["function"]["name"] == "tools__echo"
        json.loads(res.tool_calls[0]["function"]["arguments"])
    finally:
        eng.stop()


@pytest.mark.gpu
def test_bpe_serving_on_gpu_fullvocab_sampler():
    """The whole BPE serving path on hardware: HF tokenizer + token-trie
    masks + the radix-select full-vocab sampler (live vocab > 512 routes
    to csrc/sampling_fullvocab.hip) through the engine on cuda:0,
    producing a grammar-valid executable tool call."""
    import dataclasses

    from tokenizers import Tokenizer, models, pre_tokenizers, decoders, trainers

    from agentcontrolplane_amd.engine.config import PRESETS

    tok = Tokenizer(models.BPE(unk_token=None))
    tok.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tok.decoder = decoders.ByteLevel()
    trainer = trainers.BpeTrainer(
        vocab_size=2000, special_tokens=[], show_progress=False,
        min_frequency=2,
        initial_alphabet=pre_tokenizers.ByteLevel.alphabet(),
    )
    # merge-rich corpus: hundreds of distinct repeated words so the vocab
    # grows well past 512 (routing to the full-vocab sampler)
    corpus = ['{"name": "calc__add", "arguments": {"a": 1, "b": 2}}'] * 40 + [
        " ".join(f"tok{i:04d}" for i in range(j, j + 40)) for j in range(0, 800, 10)
    ] * 4
    tok.train_from_iterator(corpus, trainer)
    assert tok.get_vocab_size() > 512, tok.get_vocab_size()
    import tempfile

    with tempfile.TemporaryDirectory() as td:
        path = f"{td}/tokenizer.json"
        tok.save(path)
        PRESETS["tiny-gpu-bpe"] = dataclasses.replace(
            PRESETS["tiny-gpu"], name="tiny-gpu-bpe", vocab_size=4096
        )
        try:
            eng = InferenceEngine(
                EngineConfig(model="tiny-gpu-bpe", device="cuda",
                             num_kv_blocks=256, tokenizer_path=path,
                             max_prefill_tokens=512),
                start=True,
            )
            try:
                assert eng._live > 512  # full-vocab sampler route
                assert eng.scheduler.grammar_factory is not None
                tools = [{"type": "function", "function": {
                    "name": "calc__add",
                    "parameters": {"type": "object",
                                   "properties": {"a": {"type": "number"},
                                                  "b": {"type": "number"}},
                                   "required": ["a", "b"]}}}]
                res = eng.chat(
                    [{"role": "user", "content": "add 1 and 2"}],
                    tools=tools,
                    sampling=SamplingParams(max_tokens=64, temperature=0.9,
                                            top_p=0.95, top_k=50,
                                            tool_choice="required"),
                )
                assert res.finish_reason == "tool_calls"
                call = res.tool_calls[0]["function"]
                assert call["name"] == "calc__add"
                args = json.loads(call["arguments"])
                assert set(args) == {"a", "b"}
                # free-text turn exercises the unmasked full-vocab draw
                res2 = eng.chat(
                    [{"role": "user", "content": "hello"}],
                    sampling=SamplingParams(max_tokens=8, temperature=0.8),
                )
                assert res2.completion_tokens <= 8
            finally:
                eng.stop()
        finally:
            PRESETS.pop("tiny-gpu-bpe", None)


def test_bpe_chat_stream_deltas(bpe_path):
    """chat_stream with a BPE tokenizer: deltas arrive per decoded UTF-8
    span via token_bytes (round 1 assumed one token = one byte)."""
    eng = InferenceEngine(
        EngineConfig(model="tiny", device="cpu", num_kv_blocks=512,
                     tokenizer_path=bpe_path, max_prefill_tokens=512),
        start=True,
    )
    try:
        deltas, finals = [], []
        for kind, payload in eng.chat_stream(
            [{"role": "user", "content": "stream some text"}],
            sampling=SamplingParams(max_tokens=16, temperature=1.0),
        ):
            (deltas if kind == "delta" else finals).append(payload)
        assert finals and finals[0].completion_tokens <= 16
        joined = "".join(deltas)
        assert joined == finals[0].text  # deltas reassemble the final text
    finally:
        eng.stop()
