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
