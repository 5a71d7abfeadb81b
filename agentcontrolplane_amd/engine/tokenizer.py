"""Byte-level tokenizer + chat template.

There is no network to fetch a trained tokenizer, and the bench runs on
random-init weights, so token *identities* carry no semantics — what matters
is that (a) text→ids→text round-trips exactly, (b) the chat template
produces realistic token counts, and (c) constrained JSON decoding can map
grammar character classes onto token ids.  A byte-level vocabulary gives all
three: ids 0-255 are raw bytes; special tokens sit above.

The chat template mirrors the Llama-3 header structure:
  <|begin_of_text|> then per message
  <|start_header_id|> role-bytes <|end_header_id|> content-bytes <|eot_id|>
Tool schemas are rendered into the system message (the usual pattern for
function-calling fine-tunes).
"""
from __future__ import annotations

import json
from typing import Any, Dict, List, Sequence

BOS = 256            # <|begin_of_text|>
START_HEADER = 257   # <|start_header_id|>
END_HEADER = 258     # <|end_header_id|>
EOT = 259            # <|eot_id|> — end of message/turn
EOS = EOT
TOOL_CALL_START = 260  # <|tool_call|> sentinel opening constrained JSON
N_SPECIAL = 261


class ByteTokenizer:
    """vocab: 256 bytes + specials; everything above is dead (never sampled
    in free-run mode thanks to the sampler's vocab mask)."""

    def __init__(self, vocab_size: int):
        if vocab_size < N_SPECIAL:
            raise ValueError(f"vocab_size {vocab_size} < {N_SPECIAL}")
        self.vocab_size = vocab_size
        self.live_vocab = N_SPECIAL
        self.eot = EOT
        self.bos = BOS

    def token_bytes(self, tid: int):
        return bytes([tid]) if 0 <= tid < 256 else None

    def encode_text(self, text: str) -> List[int]:
        return list(text.encode("utf-8"))

    def decode(self, ids: Sequence[int]) -> str:
        data = bytes(i for i in ids if 0 <= i < 256)
        return data.decode("utf-8", errors="replace")

    # ------------------------------------------------------------- template

    def render_chat(
        self, messages: List[Dict[str, Any]], tools: List[Dict[str, Any]] | None = None
    ) -> List[int]:
        ids = [BOS]
        tools = tools or []
        if tools and not (messages and messages[0].get("role") == "system"):
            # no system turn to carry the tool schemas: synthesize one, else
            # the model never sees the tools (OpenAI-endpoint calls often
            # have no system message)
            messages = [{"role": "system", "content": "You can call tools."}] + list(messages)
        for i, m in enumerate(messages):
            role = m.get("role", "user")
            content = m.get("content", "")
            if i == 0 and role == "system" and tools:
                content = content + "\n\nAvailable tools:\n" + json.dumps(tools)
            ids.append(START_HEADER)
            ids.extend(self.encode_text(role))
            ids.append(END_HEADER)
            if m.get("toolCalls"):
                ids.append(TOOL_CALL_START)
                ids.extend(self.encode_text(json.dumps(m["toolCalls"])))
            else:
                ids.extend(self.encode_text(content))
            ids.append(EOT)
        # assistant generation prompt
        ids.append(START_HEADER)
        ids.extend(self.encode_text("assistant"))
        ids.append(END_HEADER)
        return ids


# --------------------------------------------------------------- BPE (HF)


def _gpt2_byte_decoder():
    """Inverse of the GPT-2 byte→unicode table that ByteLevel BPE vocabs
    are written in (the printable-alias mapping every HF byte-level
    tokenizer uses)."""
    bs = (
        list(range(ord("!"), ord("~") + 1))
        + list(range(0xA1, 0xAC + 1))
        + list(range(0xAE, 0xFF + 1))
    )
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return {chr(c): b for b, c in zip(bs, cs)}


class HFTokenizer:
    """BPE tokenizer loaded from a HuggingFace ``tokenizer.json`` — the
    serving path for real checkpoints (models/weights.py loads the
    safetensors; this loads the matching vocabulary).  Same interface as
    ByteTokenizer: ``encode_text/decode/render_chat`` plus ``live_vocab``
    (= full BPE vocab) and the special ids the engine keys on.

    Specials follow the Llama-3 naming; when the file lacks them (e.g. a
    generic BPE) they are added to the vocabulary.  ``token_bytes(tid)``
    exposes each token's raw byte string through the GPT-2 byte-level
    table — the token-trie constrained decoder (token_grammar.py) is
    built from it."""

    SPECIALS = {
        "bos": "<|begin_of_text|>",
        "start_header": "<|start_header_id|>",
        "end_header": "<|end_header_id|>",
        "eot": "<|eot_id|>",
    }

    def __init__(self, path: str, vocab_size: int | None = None):
        from tokenizers import Tokenizer

        self.tok = Tokenizer.from_file(path)
        missing = [
            name for name in self.SPECIALS.values()
            if self.tok.token_to_id(name) is None
        ]
        if missing:
            self.tok.add_special_tokens(missing)
        self.bos = self.tok.token_to_id(self.SPECIALS["bos"])
        self.start_header = self.tok.token_to_id(self.SPECIALS["start_header"])
        self.end_header = self.tok.token_to_id(self.SPECIALS["end_header"])
        self.eot = self.tok.token_to_id(self.SPECIALS["eot"])
        self.live_vocab = self.tok.get_vocab_size()
        self.vocab_size = max(vocab_size or 0, self.live_vocab)
        # token id -> raw bytes (None for specials / non-byte tokens)
        dec = _gpt2_byte_decoder()
        special_ids = {self.bos, self.start_header, self.end_header, self.eot}
        self._bytes: list = [None] * self.live_vocab
        for s, tid in self.tok.get_vocab().items():
            if tid in special_ids or (s.startswith("<|") and s.endswith("|>")):
                continue
            try:
                self._bytes[tid] = bytes(dec[c] for c in s)
            except KeyError:
                # non-byte-level vocab entry: fall back to UTF-8 of the
                # surface form (approximate, but round-trips for ASCII)
                self._bytes[tid] = s.encode("utf-8")

    # ------------------------------------------------------------ text

    def encode_text(self, text: str):
        return self.tok.encode(text, add_special_tokens=False).ids

    def decode(self, ids) -> str:
        buf = bytearray()
        for i in ids:
            b = self._bytes[i] if 0 <= i < len(self._bytes) else None
            if b is not None:
                buf.extend(b)
        return buf.decode("utf-8", errors="replace")

    def token_bytes(self, tid: int):
        """Raw bytes of one token (None for specials)."""
        return self._bytes[tid] if 0 <= tid < len(self._bytes) else None

    # -------------------------------------------------------- template

    def render_chat(self, messages, tools=None):
        """Llama-3 chat header structure, same shape as ByteTokenizer's."""
        import json as _json

        ids = [self.bos]
        tools = tools or []
        if tools and not (messages and messages[0].get("role") == "system"):
            messages = [{"role": "system", "content": "You can call tools."}] + list(messages)
        for i, m in enumerate(messages):
            role = m.get("role", "user")
            content = m.get("content", "")
            if i == 0 and role == "system" and tools:
                content = content + "\n\nAvailable tools:\n" + _json.dumps(tools)
            ids.append(self.start_header)
            ids.extend(self.encode_text(role))
            ids.append(self.end_header)
            if m.get("toolCalls"):
                ids.extend(self.encode_text(_json.dumps(m["toolCalls"])))
            else:
                ids.extend(self.encode_text(content))
            ids.append(self.eot)
        ids.append(self.start_header)
        ids.extend(self.encode_text("assistant"))
        ids.append(self.end_header)
        return ids
