"""Continuous-batching scheduler over the paged KV pool.

Every step admits (a) one decode token for every running sequence and (b)
prefill chunks from the waiting queue under a token budget
(``max_prefill_tokens``), producing one FlatBatch — the Task reconcilers'
concurrent LLM turns all land in the same forward pass (the "continuously
batch every Pending/ToolCallsPending Task" requirement of the north star).

KV pressure: when the pool cannot fit the next decode token for every
running sequence, the youngest sequences are preempted — their blocks are
freed and they re-enter the waiting queue with prompt = original prompt +
tokens generated so far (recompute-style preemption; the context window in
Task status stays the durable source of truth, the KV pool is a derived
cache).
"""
from __future__ import annotations

import dataclasses
import json
from typing import Dict, List, Optional, Tuple

import torch

from .batch import FlatBatch, SeqMeta
from .config import EngineConfig
from .grammar import ToolCallGrammar
from .request import InferenceRequest

WAITING, PREFILL, DECODE, FINISHED = "waiting", "prefill", "decode", "finished"


def worst_case_output_tokens(request: InferenceRequest) -> int:
    """Upper bound on tokens a request can emit.

    Free-running requests stop at ``max_tokens`` exactly.  Constrained
    requests can overshoot it: the grammar must finish the scaffolding
    (name + MID) and legally unwind the arguments object, so admission /
    KV reservation / context-limit checks must budget for that instead of
    trusting ``max_tokens`` (overshoot near the context limit
    walked positions past the RoPE table)."""
    sp = request.sampling
    if not request.constrained or not request.tools:
        return sp.max_tokens
    names = [t["function"]["name"] for t in request.tools]
    max_name = max(len(json.dumps(n)) - 2 for n in names)
    max_args = max(16, sp.max_tokens - 24 - max(len(n) for n in names))
    pre = 0 if getattr(request, "pre_in_prompt", False) else len(ToolCallGrammar.PRE)
    # pre + name + closing quote + MID + args + unwind slack (close braces,
    # literal completion) + EOT
    return pre + max_name + 1 + len(ToolCallGrammar.MID) + max_args + 16


class Sequence:
    __slots__ = (
        "seq_id", "request", "prompt_ids", "output_ids", "num_processed",
        "state", "grammar", "arrival",
    )

    def __init__(self, seq_id: int, request: InferenceRequest, grammar_factory=None):
        self.seq_id = seq_id
        self.request = request
        self.prompt_ids = list(request.prompt_ids)
        self.output_ids: List[int] = []
        self.num_processed = 0      # prompt tokens already in the KV cache
        self.state = WAITING
        self.grammar: Optional[ToolCallGrammar] = None
        self.arrival = request.submit_time
        if request.constrained:
            if grammar_factory is not None:
                # BPE path: token-level grammar from the engine's trie
                self.grammar = grammar_factory(request)
            else:
                names = [t["function"]["name"] for t in request.tools]
                # steer the arguments object to close before the token
                # budget: the grammar's closing mode needs ~20 tokens of
                # slack for the name/scaffolding + the shortest unwind
                max_args = max(
                    16, request.sampling.max_tokens - 24 - max(len(n) for n in names)
                )
                self.grammar = ToolCallGrammar(
                    tools=request.tools,
                    max_args_len=max_args,
                    pre_in_prompt=request.pre_in_prompt,
                )

    @property
    def total_len(self) -> int:
        return len(self.prompt_ids) + len(self.output_ids)


@dataclasses.dataclass
class SchedulerOutput:
    batch: Optional[FlatBatch]
    preempted: List[Sequence]


class Scheduler:
    def __init__(self, config: EngineConfig, block_manager, device):
        from .stage import HostStager

        self.cfg = config
        self.bm = block_manager
        self.device = torch.device(device)
        self.stager = HostStager(device)
        # set by the engine for BPE tokenizers (token-trie constrained
        # decoding); None = byte-level ToolCallGrammar
        self.grammar_factory = None
        self.waiting: List[Sequence] = []
        self.running: List[Sequence] = []   # in admission order (oldest first)
        self._by_id: Dict[int, Sequence] = {}
        self._seq_counter = 0
        # decode block-table tensor cache: rebuilt only when membership or
        # any member's block list changes (the table epoch)
        self._decode_cache_key = None
        self._decode_cache = None
        # continuation cache: finished sequences keep their KV blocks for a
        # while; a new request whose prompt extends a finished conversation
        # adopts the shared prefix instead of re-prefilling it (the agent
        # loop's next turn always extends the previous one).  KV stays a
        # derived cache — eviction only costs recompute.
        self.retired: Dict[int, tuple] = {}      # seq_id -> token tuple (FIFO)
        # longest-prefix lookup: every retiree is indexed under the rolling
        # hash of each of its block-aligned prefixes, so an adopter finds the
        # retiree with the LONGEST shared prefix in O(prompt/block) lookups
        # (a first-block bucket would keep matching a sibling's short shared
        # system prompt instead of the same conversation's previous turn)
        self._prefix_index: Dict[int, List[int]] = {}
        self._retired_hashes: Dict[int, List[int]] = {}
        self.max_retired = 4096
        self.continuation_cache = True
        self.continuation_hits = 0
        self.continuation_tokens_saved = 0

    def add_request(self, request: InferenceRequest) -> Sequence:
        self._seq_counter += 1
        seq = Sequence(self._seq_counter, request, self.grammar_factory)
        request.seq = seq
        self.waiting.append(seq)
        self._by_id[seq.seq_id] = seq
        return seq

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    def num_running(self) -> int:
        return len(self.running)

    # ---------------------------------------------------------------- step

    def schedule(self, spec_after: Optional[FlatBatch] = None) -> SchedulerOutput:
        """Build the next step's FlatBatch.

        With ``spec_after`` (the still-uncommitted previous batch), build a
        SPECULATIVE step: decode rows take their input token from the
        previous step's GPU-resident samples (``spec_src_rows``), so the
        step can launch before the previous one syncs.  Speculation bails
        (batch=None) rather than preempt, and requires every running decode
        sequence to appear in the pending step's sample rows (grammar rows
        are included — the engine defers only their sampler launch)."""
        preempted: List[Sequence] = []
        self.stager.step()  # rotate the pinned staging slots (stage.py)
        bs = self.cfg.kv_block_size

        def blocks_for(cur_len: int, have_blocks: int, n_tokens: int) -> int:
            total = (cur_len + n_tokens + bs - 1) // bs
            return max(0, total - have_blocks)

        def seq_blocks_for(s: Sequence, n_tokens: int) -> int:
            return blocks_for(self.bm.seq_len(s.seq_id), len(self.bm.block_table(s.seq_id)), n_tokens)

        # 1) reserve one decode token per running decode sequence; preempt
        #    the youngest until the reservations fit the pool
        spec_src: Dict[int, int] = {}
        if spec_after is not None:
            for i, sid in enumerate(spec_after.sample_seq_ids):
                spec_src[sid] = i
            decode_seqs = []
            for s in self.running:
                if s.state != DECODE:
                    continue
                if s.seq_id not in spec_src:
                    return SchedulerOutput(batch=None, preempted=preempted)
                # grammar-constrained rows ARE speculable: the forward does
                # not depend on the pending token, only the sampler's mask
                # does — the engine defers the sampler launch of a
                # grammar-carrying speculative step until the predecessor's
                # commit has advanced the grammar states.
                cap = s.request.sampling.max_tokens * (8 if s.grammar is not None else 1)
                if len(s.request.output_ids) + 2 > cap:
                    continue  # finishes by length at the pending commit
                decode_seqs.append(s)
            if not decode_seqs and not self.waiting and not any(
                s.state == PREFILL for s in self.running
            ):
                return SchedulerOutput(batch=None, preempted=preempted)
        else:
            decode_seqs = [s for s in self.running if s.state == DECODE]
        while decode_seqs:
            decode_need = sum(seq_blocks_for(s, 1) for s in decode_seqs)
            if decode_need <= self.bm.free_blocks:
                break
            if self._evict_one_retired():
                continue  # reclaimed cache blocks before preempting live work
            if spec_after is not None:
                return SchedulerOutput(batch=None, preempted=preempted)
            victim = decode_seqs.pop()  # youngest
            self.running.remove(victim)
            self._preempt(victim)
            preempted.append(victim)
        avail = self.bm.free_blocks - sum(seq_blocks_for(s, 1) for s in decode_seqs)

        def fit_chunk(cur_len: int, have_blocks: int, want: int) -> int:
            """Largest chunk ≤ want that fits in `avail` new blocks."""
            slack = have_blocks * bs - cur_len  # room in the last partial block
            return min(want, slack + avail * bs)

        # 2) admit prefill chunks under the token budget and block budget
        budget = self.cfg.max_prefill_tokens
        prefills: List[Tuple[Sequence, int]] = []
        # continue partially-prefilled running sequences first
        for s in self.running:
            if s.state == PREFILL and budget > 0:
                want = min(len(s.prompt_ids) - s.num_processed, budget)
                cur = self.bm.seq_len(s.seq_id)
                have = len(self.bm.block_table(s.seq_id))
                chunk = fit_chunk(cur, have, want)
                if chunk <= 0 and want > 0 and self.retired:
                    # the continuation cache must never starve live work:
                    # reclaim retirees for a running prefill exactly as the
                    # admission path does for waiting sequences
                    before = self.bm.free_blocks
                    self._reclaim(self.bm.free_blocks + (want + bs - 1) // bs)
                    avail += self.bm.free_blocks - before
                    chunk = fit_chunk(cur, have, want)
                if chunk > 0:
                    prefills.append((s, chunk))
                    budget -= chunk
                    avail -= blocks_for(cur, have, chunk)
        # then admit waiting sequences
        while self.waiting and budget > 0 and len(self.running) < self.cfg.max_batch_size:
            s = self.waiting[0]
            total_blocks = (
                s.total_len + worst_case_output_tokens(s.request) + bs - 1
            ) // bs
            if total_blocks > self.bm.num_blocks:
                self.waiting.pop(0)
                self.abort(
                    s,
                    ValueError(
                        f"request needs {total_blocks} KV blocks but the pool has "
                        f"{self.bm.num_blocks}"
                    ),
                )
                continue
            want0 = min(len(s.prompt_ids), budget)
            if avail * bs < want0 and self.retired:
                before = self.bm.free_blocks
                self._reclaim(self.bm.free_blocks + (want0 + bs - 1) // bs)
                avail += self.bm.free_blocks - before
            if fit_chunk(0, 0, 1) <= 0:
                break
            self.waiting.pop(0)
            adopted = self._try_adopt(s)
            if not adopted:
                self.bm.add_seq(s.seq_id)
                s.num_processed = 0
            cur = self.bm.seq_len(s.seq_id)
            have = len(self.bm.block_table(s.seq_id))
            want = min(len(s.prompt_ids) - s.num_processed, budget)
            chunk = fit_chunk(cur, have, want)
            if chunk <= 0:
                # adopted blocks but no room to extend this step: keep it
                # running as a pending prefill for the next step
                s.state = PREFILL
                self.running.append(s)
                continue
            s.state = PREFILL
            self.running.append(s)
            prefills.append((s, chunk))
            budget -= chunk
            avail -= blocks_for(cur, have, chunk)

        if spec_after is None:
            decode_seqs = [s for s in self.running if s.state == DECODE]
        if not prefills and not decode_seqs:
            # starvation break: running PREFILL sequences exist but none
            # could take a chunk (pool exhausted mid-prefill).  Decode
            # sequences are preempted above, but two-plus concurrent
            # prefills can jointly exhaust the pool with no decode victim
            # — admission bounds each SINGLE sequence to fit the pool, so
            # preempting the youngest stalled prefill (recompute-style)
            # guarantees the oldest eventually completes.  Folding makes
            # this reachable in practice (folded sequences re-enter
            # prefill holding their KV).
            if spec_after is None:
                stalled = [s for s in self.running if s.state == PREFILL]
                if len(stalled) > 1 and budget > 0:
                    victim = stalled[-1]  # youngest
                    self.running.remove(victim)
                    self._preempt(victim)
                    preempted.append(victim)
            return SchedulerOutput(batch=None, preempted=preempted)

        # 3) materialize the flat batch
        token_ids: List[int] = []
        positions: List[int] = []
        slot_mapping: List[int] = []
        prefill_metas: List[SeqMeta] = []
        logit_rows: List[int] = []
        sample_seq_ids: List[int] = []
        row = 0
        for s, chunk in prefills:
            start = s.num_processed
            toks = s.prompt_ids[start : start + chunk]
            slots = self.bm.append_tokens(s.seq_id, chunk)
            token_ids.extend(toks)
            positions.extend(range(start, start + chunk))
            slot_mapping.extend(slots)
            s.num_processed += chunk
            done = s.num_processed == len(s.prompt_ids)
            prefill_metas.append(
                SeqMeta(
                    seq_id=s.seq_id,
                    query_len=chunk,
                    seq_len=s.num_processed,
                    ctx_len=start,
                    block_table=list(self.bm.block_table(s.seq_id)),
                    needs_logits=done,
                )
            )
            if done:
                logit_rows.append(row + chunk - 1)
                sample_seq_ids.append(s.seq_id)
                s.state = DECODE
            row += chunk
        num_prefill_tokens = row

        decode_block_tables = None
        decode_seq_lens = None
        decode_tables_i32 = None
        decode_ids: List[int] = []
        spec_rows: List[int] = []
        if decode_seqs:
            lens = []
            for s in decode_seqs:
                if spec_after is not None:
                    tok = 0  # filled on device from the previous step's samples
                    pos = s.total_len  # the in-flight token occupies total_len
                    spec_rows.append(spec_src[s.seq_id])
                else:
                    tok = s.output_ids[-1] if s.output_ids else s.prompt_ids[-1]
                    pos = s.total_len - 1
                slots = self.bm.append_tokens(s.seq_id, 1)
                token_ids.append(tok)
                positions.append(pos)
                slot_mapping.extend(slots)
                lens.append(pos + 1)
                logit_rows.append(row)
                sample_seq_ids.append(s.seq_id)
                decode_ids.append(s.seq_id)
                row += 1
            # block tables change only when membership changes or a table
            # grows (one block per kv_block_size steps): cache the tensor
            key = (tuple(decode_ids), self.bm.table_epoch)
            if key == self._decode_cache_key:
                decode_block_tables, decode_tables_i32 = self._decode_cache
            else:
                import numpy as np

                tables = [self.bm.block_table(i) for i in decode_ids]
                max_blocks = max(len(t) for t in tables)
                padded = [t + [0] * (max_blocks - len(t)) for t in tables]
                decode_block_tables = self.stager.fresh(padded, np.int64)
                decode_tables_i32 = decode_block_tables.int()
                self._decode_cache_key = key
                self._decode_cache = (decode_block_tables, decode_tables_i32)
                self._decode_cache_host = padded
            decode_seq_lens = self.stager.tensor("dlens", lens, "int64")

        batch = FlatBatch(
            token_ids=self.stager.tensor("tok", token_ids, "int64"),
            positions=self.stager.tensor("pos", positions, "int64"),
            slot_mapping=self.stager.tensor("slot", slot_mapping, "int64"),
            prefills=prefill_metas,
            num_prefill_tokens=num_prefill_tokens,
            decode_seq_ids=decode_ids,
            decode_block_tables=decode_block_tables,
            decode_seq_lens=decode_seq_lens,
            logit_rows=self.stager.tensor("lrow", logit_rows, "int64"),
            sample_seq_ids=sample_seq_ids,
        )
        batch._stager = self.stager
        # host-side copies for the TP wire broadcast (batch_to_wire):
        # avoids per-step D2H tolist() syncs on rank 0
        batch._host = {
            "token_ids": token_ids,
            "positions": positions,
            "slot_mapping": slot_mapping,
            "logit_rows": logit_rows,
            "decode_block_tables": getattr(self, "_decode_cache_host", None)
            if decode_block_tables is not None else None,
            "decode_seq_lens": lens if decode_seqs else None,
        }
        if decode_tables_i32 is not None:
            batch._decode_tables_i32 = decode_tables_i32
        if spec_after is not None and spec_rows:
            batch.spec_src_rows = self.stager.tensor("spec", spec_rows, "int64")
        return SchedulerOutput(batch=batch, preempted=preempted)

    # ------------------------------------------------------------ commit

    def seq_by_id(self, seq_id: int) -> Optional[Sequence]:
        return self._by_id.get(seq_id)

    def append_sampled(self, seq: Sequence, token: int) -> None:
        seq.output_ids.append(token)
        seq.request.output_ids.append(token)

    def finish_seq(self, seq: Sequence, reason: str) -> None:
        seq.state = FINISHED
        if seq in self.running:
            self.running.remove(seq)
        if self.bm.has_seq(seq.seq_id):
            if self.continuation_cache and self.bm.seq_len(seq.seq_id) >= self.cfg.kv_block_size:
                self._retire(seq)
            else:
                self.bm.free_seq(seq.seq_id)
        self._by_id.pop(seq.seq_id, None)
        seq.request.finish(reason)

    # ------------------------------------------------------- continuation

    def _prefix_hashes(self, tokens, max_blocks: Optional[int] = None) -> List[int]:
        """Rolling hashes of tokens[:bs], tokens[:2*bs], … (full blocks)."""
        bs = self.cfg.kv_block_size
        n_blocks = len(tokens) // bs
        if max_blocks is not None:
            n_blocks = min(n_blocks, max_blocks)
        out: List[int] = []
        h = 0
        for b in range(n_blocks):
            h = hash((h, tuple(tokens[b * bs : (b + 1) * bs])))
            out.append(h)
        return out

    def _retire(self, seq: Sequence) -> None:
        tokens = tuple(seq.prompt_ids + seq.output_ids)
        cached = self.bm.seq_len(seq.seq_id)
        tokens = tokens[:cached]
        hashes = self._prefix_hashes(tokens)
        if not hashes:
            self.bm.free_seq(seq.seq_id)
            return
        self.retired[seq.seq_id] = tokens
        self._retired_hashes[seq.seq_id] = hashes
        for h in hashes:
            self._prefix_index.setdefault(h, []).append(seq.seq_id)
        while len(self.retired) > self.max_retired:
            self._evict_one_retired()

    def _unindex(self, old_id: int) -> None:
        for h in self._retired_hashes.pop(old_id, []):
            bucket = self._prefix_index.get(h)
            if bucket is not None:
                try:
                    bucket.remove(old_id)
                except ValueError:
                    pass
                if not bucket:
                    self._prefix_index.pop(h, None)

    def _evict_one_retired(self) -> bool:
        if not self.retired:
            return False
        old_id = next(iter(self.retired))
        self.retired.pop(old_id)
        self._unindex(old_id)
        self.bm.free_seq(old_id)
        return True

    def _reclaim(self, need_blocks: int) -> None:
        while self.bm.free_blocks < need_blocks and self._evict_one_retired():
            pass

    def _try_adopt(self, seq: Sequence) -> bool:
        """Share the retired conversation with the LONGEST matching KV
        prefix (refcounted, full blocks only); returns True if the sequence
        was registered with the block manager.  Retirees stay cached so any
        number of requests reuse them — the same task's next turn (long
        match) or sibling tasks sharing a system prompt (short match)."""
        bs = self.cfg.kv_block_size
        if not self.continuation_cache or len(seq.prompt_ids) <= bs:
            return False
        # at least one token must remain to prefill (logits come from it)
        max_blocks = (len(seq.prompt_ids) - 1) // bs
        hashes = self._prefix_hashes(seq.prompt_ids, max_blocks)
        best_id, n_blocks = None, 0
        for b in range(len(hashes) - 1, -1, -1):
            for cand in self._prefix_index.get(hashes[b], ()):  # longest first
                toks = self.retired[cand]
                want = (b + 1) * bs
                if len(toks) >= want and toks[:want] == tuple(seq.prompt_ids[:want]):
                    best_id, n_blocks = cand, b + 1
                    break
            if best_id is not None:
                break
        if best_id is None or n_blocks < 1:
            return False
        self.bm.share_prefix(seq.seq_id, best_id, n_blocks, n_blocks * bs)
        seq.num_processed = n_blocks * bs
        self.continuation_hits += 1
        self.continuation_tokens_saved += n_blocks * bs
        return True

    def _preempt(self, seq: Sequence) -> None:
        """Recompute-style preemption: blocks freed, prompt grows to include
        generated tokens, back to the head of the waiting queue."""
        if self.bm.has_seq(seq.seq_id):
            self.bm.free_seq(seq.seq_id)
        seq.prompt_ids = seq.prompt_ids + seq.output_ids
        # keep outputs in the request — generation continues from here
        seq.output_ids = []
        seq.num_processed = 0
        seq.state = WAITING
        self.waiting.insert(0, seq)

    def cancel_request(self, request) -> None:
        """Stop generating for a request (e.g. a streaming client went
        away).  Runs in the engine loop thread; a finished/unknown request
        is a no-op.  The sequence finishes with reason "cancelled" — its KV
        retires into the continuation cache like any other finish."""
        seq = request.seq
        if seq is None or seq.state == FINISHED:
            return
        if seq in self.waiting:
            self.waiting.remove(seq)
        self.finish_seq(seq, "cancelled")

    def abort(self, seq: Sequence, err: BaseException) -> None:
        seq.state = FINISHED
        if seq in self.running:
            self.running.remove(seq)
        if seq in self.waiting:
            self.waiting.remove(seq)
        if self.bm.has_seq(seq.seq_id):
            self.bm.free_seq(seq.seq_id)
        self._by_id.pop(seq.seq_id, None)
        seq.request.fail(err)
