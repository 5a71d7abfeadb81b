"""Token-level constrained decoding over a BPE vocabulary.

The byte-level PDA (grammar.py) constrains one byte per step — exactly one
token with ByteTokenizer.  A BPE tokenizer emits multi-byte tokens, so the
per-step mask must contain every token whose ENTIRE byte string the PDA
accepts from its current state.  That set comes from a depth-first walk of
a byte trie over the vocabulary (built once per tokenizer): an edge is
taken only when its byte is in the automaton's allowed set for the state
reached so far, advancing a cloned automaton along the path; every trie
node holding a token id contributes that token.

Walk cost is bounded by the number of legal prefixes, which for JSON
string interiors can still be large — masks are therefore memoized on a
structural state signature (PDA state + stack + phase, EXCLUDING length
counters while far from the budget, so signatures repeat across steps and
sequences and the steady-state cost is a dict lookup).

Cited reference behavior: none — the reference offloads validity to the
remote provider; this is the MI355X-native replacement the north star
names ("tool-call JSON constrained decoding" on the hot path).
"""
from __future__ import annotations

import copy
from typing import Dict, List, Optional, Set, Tuple

from .grammar import ToolCallGrammar


class TokenTrie:
    """Byte trie over the vocabulary.  Node = [children: dict byte->node,
    token_ids: list].  Multiple ids can share one byte string (duplicate
    surface forms)."""

    __slots__ = ("root", "n_tokens")

    def __init__(self, token_bytes: List[Optional[bytes]]):
        self.root = ({}, [])
        self.n_tokens = len(token_bytes)
        for tid, bs in enumerate(token_bytes):
            if not bs:
                continue
            node = self.root
            for b in bs:
                nxt = node[0].get(b)
                if nxt is None:
                    nxt = ({}, [])
                    node[0][b] = nxt
                node = nxt
            node[1].append(tid)


def _clone(g):
    """Cheap structural clone of a grammar/PDA; grammar classes provide
    targeted clone() (deepcopy per trie edge would dominate the walk)."""
    c = getattr(g, "clone", None)
    return c() if c is not None else copy.deepcopy(g)


class TokenGrammar:
    """Drop-in for ToolCallGrammar at the token level: ``allowed_tokens()``
    returns legal TOKEN ids, ``advance(tid)`` consumes one token (advancing
    the byte PDA through the token's bytes), ``accepting``/``parse()``
    delegate to the underlying byte grammar."""

    def __init__(self, trie: TokenTrie, token_bytes: List[Optional[bytes]],
                 eot: int, tools: List[dict], max_args_len: int = 2048,
                 pre_in_prompt: bool = False,
                 mask_cache: Optional[Dict] = None):
        self.trie = trie
        self.token_bytes = token_bytes
        self.eot = eot
        self.byte_grammar = ToolCallGrammar(
            tools=tools, max_args_len=max_args_len, pre_in_prompt=pre_in_prompt
        )
        # shared across sequences with the same tokenizer+tools (the
        # engine passes one per (trie, tools-signature))
        self._mask_cache = mask_cache if mask_cache is not None else {}

    # ----------------------------------------------------------- signature

    def _signature(self) -> Optional[Tuple]:
        """Structural state key for memoization, or None when near a length
        budget (closing mode makes allowed sets depend on the counters).
        Every key is prefixed with the tool-set identity and the selected
        name: a multi-byte token can span phase boundaries (e.g. cover the
        tail of PRE plus the start of a tool name), so masks are only
        shareable between grammars constraining the same tools."""
        g = self.byte_grammar
        pre = (tuple(g.names), bytes(g.name_prefix))
        if g.phase == "pre":
            return pre + ("pre", g.pos)
        if g.phase == "name":
            return pre + ("name",)
        if g.phase == "mid":
            return pre + ("mid", g.pos)
        if g.phase in ("done", "close"):
            return pre + (g.phase,)
        # args: depends on the machine
        a = g.args
        if a.__class__.__name__ == "JsonValueMachine":
            if a.count > a.max_len - 64:
                return None  # closing mode approaching: exact walk
            return pre + ("json", a.state, tuple(a.stack), a.done)
        # SchemaArgsMachine
        if a.count > a.max_len - 64:
            return None
        sub_sig = None
        if a.sub is not None:
            if a.sub.count > a.sub.max_len - 64:
                return None
            sub_sig = (a.sub.state, tuple(a.sub.stack), a.sub.done)
        return pre + (
            "schema", a.pi, a.off, a.u8, a.u8lo, a.u8hi, a._bool_rest, sub_sig,
            tuple(a.fields),
        )

    # --------------------------------------------------------------- masks

    def allowed_tokens(self) -> Set[int]:
        sig = self._signature()
        if sig is not None:
            hit = self._mask_cache.get(sig)
            if hit is not None:
                return hit
        out = self._walk()
        if sig is not None:
            self._mask_cache[sig] = out
        return out

    def _walk(self) -> Set[int]:
        """DFS of the trie pruned by the byte PDA.  A token is legal iff
        every byte of it advances the automaton; tokens that would run past
        the accepting state are pruned naturally (phase "done" allows only
        EOT, which is never a trie byte)."""
        out: Set[int] = set()
        # iterative DFS: (node, grammar-at-node); the live grammar at the
        # root is only read, never advanced
        stack = [(self.trie.root, self.byte_grammar)]
        while stack:
            node, g = stack.pop()
            allowed = g.allowed_tokens()
            children = node[0]
            if len(children) < len(allowed):
                it = ((b, children[b]) for b in children if b in allowed)
            else:
                it = ((b, children[b]) for b in allowed if b in children)
            for b, child in it:
                g2 = _clone(g)
                g2.advance(b)
                if child[1]:
                    out.update(child[1])
                stack.append((child, g2))
        return out

    # ------------------------------------------------------------- protocol

    def advance(self, tid: int) -> None:
        if tid == self.eot:
            self.byte_grammar.advance(self.byte_grammar_eot())
            return
        bs = self.token_bytes[tid] if 0 <= tid < len(self.token_bytes) else None
        if not bs:
            raise ValueError(f"token {tid} has no byte form (special?)")
        for b in bs:
            self.byte_grammar.advance(b)

    def byte_grammar_eot(self) -> int:
        from .tokenizer import EOT

        return EOT

    def forced_run(self, cap: int = 64):
        """No byte-level folding for BPE: multi-byte tokens already cover
        forced scaffolding in a handful of steps, and folding would need
        token-boundary alignment of the forced byte string."""
        return []

    @property
    def phase(self) -> str:
        return self.byte_grammar.phase

    @property
    def accepting(self) -> bool:
        return self.byte_grammar.accepting

    @property
    def finished(self) -> bool:
        return self.byte_grammar.finished

    def parse(self):
        return self.byte_grammar.parse()
