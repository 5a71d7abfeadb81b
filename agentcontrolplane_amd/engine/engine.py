"""The inference engine: model + paged KV + continuous batching loop.

This replaces the reference's remote-LLM HTTP path (llmclient/
langchaingo_client.go): ``chat()`` turns an ACP context window into token
ids, submits an InferenceRequest, and blocks on its future while the engine
thread continuously batches every in-flight request into shared GPU steps.

Sampling is over the tokenizer's live vocabulary — bytes + specials for
the synthetic byte tokenizer, the full BPE vocabulary when a
tokenizer.json is loaded (the ≤512-entry bitonic sampler vs the
radix-select full-vocab kernel route on that size); the full-vocab
LM-head GEMM is always computed (that cost is real and stays in the
measured path).  Constrained (tool-call) sequences additionally mask to
the grammar's legal next tokens each step — single bytes under the byte
PDA, whole tokens under the token-trie walk (token_grammar.py) — and
runs of forced scaffolding bytes fold into one prefill chunk
(_fold_forced) instead of sequential masked decode steps.
"""
from __future__ import annotations

import threading
import time
from typing import Any, Dict, List, Optional

import torch

from .. import ops
from ..models import create_model
from .config import EngineConfig
from .kv import make_block_manager
from .request import ChatResult, InferenceRequest, SamplingParams
from .scheduler import Scheduler, Sequence
from .tokenizer import N_SPECIAL, ByteTokenizer


class InferenceEngine:
    def __init__(self, config: EngineConfig, model=None, start: bool = True):
        self.cfg = config
        self.mcfg = config.model_config()
        self.device = torch.device(config.device)
        if self.device.type == "cuda":
            self._load_gemm_tunings()
        import os as _os

        tok_path = config.tokenizer_path
        if tok_path is None and config.checkpoint_path:
            cand = _os.path.join(config.checkpoint_path, "tokenizer.json")
            if _os.path.exists(cand):
                tok_path = cand
        if tok_path:
            from .tokenizer import HFTokenizer

            self.tokenizer = HFTokenizer(tok_path, self.mcfg.vocab_size)
        else:
            self.tokenizer = ByteTokenizer(self.mcfg.vocab_size)
        self._eot = self.tokenizer.eot
        self._live = min(self.tokenizer.live_vocab, self.mcfg.vocab_size)
        self.tp_world = config.tensor_parallel
        self.tp_rank = 0
        if model is not None:
            self.model = model
        else:
            if self.tp_world > 1:
                import torch.distributed as dist

                assert dist.is_initialized(), "tensor_parallel > 1 needs torch.distributed"
                self.tp_rank = dist.get_rank()
                from ..parallel.tp import make_all_reduce

                self.model = create_model(
                    self.mcfg, config, config.device,
                    tp_rank=self.tp_rank, tp_world=self.tp_world,
                )
                self._init_weights()
                self.model.all_reduce = make_all_reduce()
            else:
                self.model = create_model(self.mcfg, config, config.device)
                self._init_weights()
        num_blocks = self._size_kv_pool()
        # +1 scratch block: hipGraph decode padding rows write their KV there
        self.model.allocate_kv_cache(num_blocks + 1, config.kv_block_size)
        self.bm = make_block_manager(
            num_blocks, config.kv_block_size, prefer_native=True
        )
        self.scheduler = Scheduler(config, self.bm, self.device)
        if getattr(self.tokenizer, "token_bytes", None) and self._live > N_SPECIAL:
            # BPE tokenizer: token-trie constrained decoding (token_grammar)
            from .token_grammar import TokenGrammar, TokenTrie

            tb = [self.tokenizer.token_bytes(i) for i in range(self._live)]
            trie = TokenTrie(tb)
            mask_cache: Dict = {}

            def _grammar_factory(request):
                max_args = min(2048, max(32, request.sampling.max_tokens * 4))
                return TokenGrammar(
                    trie, tb, self._eot, request.tools, max_args_len=max_args,
                    pre_in_prompt=request.pre_in_prompt, mask_cache=mask_cache,
                )

            self.scheduler.grammar_factory = _grammar_factory
        self.graph_runner = None
        # MoE decode batches under dense_moe_threshold take the all-experts
        # dense path (static control flow) and are capture-safe; larger MoE
        # batches (sparse routing loop) and EP dispatch stay eager
        moe_graph_cap = None
        if self.mcfg.is_moe:
            if self.tp_world > 1 or getattr(self.model, "moe_dispatch", None) is not None:
                moe_graph_cap = 0
            else:
                moe_graph_cap = getattr(self.model, "dense_moe_threshold", 0)
        graphs_on = (
            self.device.type == "cuda"
            and not config.enforce_eager
            and (moe_graph_cap is None or moe_graph_cap > 0)
        )
        if graphs_on:
            from .graphs import DecodeGraphRunner

            max_blocks_per_seq = (
                min(config.max_model_len, self.mcfg.max_position)
                + config.kv_block_size - 1
            ) // config.kv_block_size
            # 1024 covers the decode-only tail of the 1k-concurrent-Task
            # bench (512 left ~half those steps on the eager path)
            max_graph_batch = min(config.max_batch_size, 1024)
            if moe_graph_cap is not None:
                max_graph_batch = min(max_graph_batch, moe_graph_cap)
            self.graph_runner = DecodeGraphRunner(
                self.model,
                max_batch=max_graph_batch,
                max_blocks_per_seq=max_blocks_per_seq,
                scratch_block=num_blocks,
                kv_block_size=config.kv_block_size,
            )
        self._ctx_limit = min(config.max_model_len, self.mcfg.max_position)
        self._gen = torch.Generator(device=self.device.type)
        self._gen.manual_seed(config.seed)
        self._sampling_cache_key = None
        self._sampling_cache = None
        # persistent pinned→device staging for per-step sampling metadata
        # (see stage.PersistentStage: unpinned rebuilds were ~10 ms/step)
        import numpy as _np

        cap = config.max_batch_size + 8
        stager = self.scheduler.stager
        self._ps_temp = stager.persistent(cap, (), _np.float32)
        self._ps_topk = stager.persistent(cap, (), _np.int64)
        self._ps_topp = stager.persistent(cap, (), _np.float32)
        if self._live <= 512:
            # byte vocab: dense per-row masks (bitonic sampler)
            self._ps_mask = stager.persistent(cap, (self._live,), _np.bool_)
            self._ps_maskmap = None
            self._mask_rows = cap
        else:
            # BPE vocab: compact mask rows + per-row indirection (a dense
            # [B, 128k] pinned mask would be hundreds of MB); rows beyond
            # the cap take a slow in-place masked_fill path
            self._mask_rows = 128
            self._ps_mask = stager.persistent(self._mask_rows, (self._live,), _np.bool_)
            self._ps_maskmap = stager.persistent(cap, (), _np.int32)
        self._spec_bail = False  # one-shot: set by a fold-pending commit
        self._lock = threading.Lock()
        self._work = threading.Condition(self._lock)
        self._pending: List[InferenceRequest] = []
        self._cancelled: List[InferenceRequest] = []
        self._running = False
        self._thread: Optional[threading.Thread] = None
        # metrics
        self._m = {
            "steps": 0,
            "prompt_tokens": 0,
            "generated_tokens": 0,
            "preemptions": 0,
            "requests_completed": 0,
            "busy_time_s": 0.0,
            "sched_time_s": 0.0,
            "compute_time_s": 0.0,
            "sample_time_s": 0.0,
        }
        self._start_time = time.monotonic()
        if start:
            self.start()

    def _load_gemm_tunings(self) -> None:
        """Load the committed TunableOp solution table (tools/tune_gemms.py)
        so library GEMMs use the autotuned hipBLASLt/rocBLAS kernels
        (+23% decode throughput vs heuristic selection).  Tuning itself
        stays off — unknown shapes fall back to the heuristic."""
        import os

        path = os.path.join(
            os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
            "..", "tuned", f"gemm_{self.cfg.model}.csv",
        )
        path = os.path.normpath(path)
        if not os.path.exists(path):
            return
        try:
            torch.cuda.tunable.enable(True)
            torch.cuda.tunable.tuning_enable(False)
            torch.cuda.tunable.read_file(path)
        except Exception as e:  # pragma: no cover
            print(f"[engine] TunableOp load failed ({e}); using heuristics")

    def _init_weights(self) -> None:
        if self.cfg.checkpoint_path:
            from ..models.weights import load_checkpoint

            load_checkpoint(self.model, self.cfg.checkpoint_path)
        else:
            self.model.random_init(self.cfg.seed)

    # ------------------------------------------------------------- sizing

    def _size_kv_pool(self) -> int:
        if self.cfg.num_kv_blocks is not None:
            return self.cfg.num_kv_blocks
        if self.device.type != "cuda":
            return 1024
        free, total = torch.cuda.mem_get_info(self.device)
        budget = int(total * self.cfg.gpu_memory_utilization) - (total - free)
        block_bytes = self.model.kv_block_bytes(self.cfg.kv_block_size)
        # keep headroom for activations: 4 GB or 5% of total
        headroom = max(4 << 30, int(0.05 * total))
        n = max(64, (budget - headroom) // block_bytes)
        return int(n)

    # ----------------------------------------------------------- lifecycle

    def start(self) -> None:
        if self._running:
            return
        self._running = True
        self._thread = threading.Thread(target=self._loop, name="acp-engine", daemon=True)
        self._thread.start()

    def stop(self) -> None:
        with self._work:
            self._running = False
            self._work.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None
        if self.tp_world > 1 and self.tp_rank == 0:
            try:
                from ..parallel.tp import STOP, broadcast_step

                broadcast_step(STOP)
            except Exception:
                pass

    # ------------------------------------------------------------- submit

    def submit(self, request: InferenceRequest) -> InferenceRequest:
        from .scheduler import worst_case_output_tokens

        limit = min(self.cfg.max_model_len, self.mcfg.max_position)
        worst = worst_case_output_tokens(request)
        if len(request.prompt_ids) + worst > limit:
            raise ValueError(
                f"request length {len(request.prompt_ids)}+{worst} (worst-case "
                f"emission) exceeds the model context limit {limit}"
            )
        with self._work:
            if len(self._pending) >= self.cfg.max_queue:
                raise RuntimeError("engine queue full")
            self._pending.append(request)
            self._work.notify_all()
        return request

    def cancel(self, req: InferenceRequest) -> None:
        """Stop generating for ``req`` at the next step boundary (used when
        a streaming client disconnects).  Safe from any thread."""
        with self._work:
            self._cancelled.append(req)
            self._work.notify_all()

    def generate(
        self,
        prompt_ids: List[int],
        sampling: Optional[SamplingParams] = None,
        constrained: bool = False,
        tools: Optional[List[Dict]] = None,
        timeout: Optional[float] = None,
    ) -> InferenceRequest:
        req = InferenceRequest(prompt_ids, sampling or SamplingParams(), constrained, tools)
        self.submit(req)
        req.wait(timeout or self.cfg.request_timeout_s)
        return req

    def _build_request(self, messages, tools, sampling) -> InferenceRequest:
        sampling = sampling or SamplingParams()
        tools = tools or []
        prompt_ids = self.tokenizer.render_chat(messages, tools)
        constrained = False
        if tools:
            choice = sampling.tool_choice
            if choice == "auto":
                # synthetic-workload policy (random-init weights cannot decide):
                # call a tool once per conversation, then answer.  A trained
                # checkpoint resolves "auto" from the TOOL_CALL_START logit.
                has_tool_result = any(m.get("role") == "tool" for m in messages)
                constrained = not has_tool_result
            elif choice == "required":
                constrained = True
        pre_in_prompt = False
        if constrained:
            # the tool-call JSON always opens with the forced PRE bytes:
            # emitting them as prompt tokens turns len(PRE) sequential
            # masked decode steps into one chunked-prefill extension
            from .grammar import ToolCallGrammar

            prompt_ids = prompt_ids + self.tokenizer.encode_text(
                ToolCallGrammar.PRE.decode()
            )
            pre_in_prompt = True
        return InferenceRequest(prompt_ids, sampling, constrained, tools, pre_in_prompt)

    def _result_of(self, req: InferenceRequest, t0: float) -> ChatResult:
        out_ids = req.output_ids
        result = ChatResult(
            prompt_tokens=len(req.prompt_ids),
            completion_tokens=len(out_ids),
            finish_reason=req.finish_reason,
            latency_s=time.monotonic() - t0,
        )
        if req.finish_reason == "tool_calls" and req.seq is not None and req.seq.grammar:
            name, args = req.seq.grammar.parse()
            result.tool_calls = [
                {
                    "id": f"call_{req.request_id:08d}",
                    "type": "function",
                    "function": {"name": name, "arguments": args},
                }
            ]
        else:
            result.text = self.tokenizer.decode(out_ids)
        return result

    def chat(
        self,
        messages: List[Dict[str, Any]],
        tools: Optional[List[Dict[str, Any]]] = None,
        sampling: Optional[SamplingParams] = None,
    ) -> ChatResult:
        """The LLMClient.send_request counterpart (blocking)."""
        t0 = time.monotonic()
        req = self._build_request(messages, tools, sampling)
        self.submit(req)
        req.wait(self.cfg.request_timeout_s)
        return self._result_of(req, t0)

    def chat_stream(
        self,
        messages: List[Dict[str, Any]],
        tools: Optional[List[Dict[str, Any]]] = None,
        sampling: Optional[SamplingParams] = None,
    ):
        """Streaming chat: yields ("delta", text) per decoded UTF-8 span as
        tokens commit, then exactly one ("done", ChatResult).  Byte tokens
        are buffered until they form valid UTF-8 (multi-byte codepoints
        split across tokens).  Constrained (tool-call) turns yield no
        deltas — the tool_calls arrive on the final result."""
        import queue as _queue

        t0 = time.monotonic()
        req = self._build_request(messages, tools, sampling)
        q: "_queue.Queue" = _queue.Queue()
        if not req.constrained:
            req.on_token = q.put
        req.on_complete = lambda r: q.put(None)
        self.submit(req)
        buf = bytearray()
        deadline = t0 + self.cfg.request_timeout_s
        try:
            while True:
                try:
                    tok = q.get(timeout=max(0.0, deadline - time.monotonic()))
                except _queue.Empty:
                    raise TimeoutError(f"request {req.request_id} timed out") from None
                if tok is None:
                    break
                tb = self.tokenizer.token_bytes(tok)
                if tb:
                    buf.extend(tb)
                    try:
                        text = buf.decode("utf-8")
                    except UnicodeDecodeError:
                        continue  # mid-codepoint: wait for the next byte
                    buf.clear()
                    yield ("delta", text)
        finally:
            # the consumer may abandon the stream (client disconnect →
            # GeneratorExit): stop generating instead of running to
            # max_tokens
            if not req._event.is_set():
                self.cancel(req)
        if req.error is not None:
            raise req.error
        if buf:
            yield ("delta", buf.decode("utf-8", errors="replace"))
        yield ("done", self._result_of(req, t0))

    def chat_async(self, messages, tools, sampling, callback) -> InferenceRequest:
        """Submit a chat turn; ``callback(result, error)`` fires from the
        engine thread on completion.  This is the serving path: reconciler
        workers never block on a turn."""
        t0 = time.monotonic()
        req = self._build_request(messages, tools, sampling)

        def on_complete(r: InferenceRequest) -> None:
            try:
                if r.error is not None:
                    callback(None, r.error)
                    return
                try:
                    result = self._result_of(r, t0)
                except Exception as e:  # result extraction failed: the
                    # request must still complete (a swallowed exception
                    # here left the Task hanging to its timeout)
                    callback(None, e)
                    return
                callback(result, None)
            except Exception:  # noqa: BLE001 — callback bugs must not kill the loop
                import traceback

                traceback.print_exc()

        req.on_complete = on_complete
        try:
            self.submit(req)
        except Exception as e:
            callback(None, e)
        return req

    # ------------------------------------------------------------ the loop

    def _loop(self) -> None:
        pending = None  # launched-but-uncommitted step
        spec_ok = self.cfg.async_scheduling and self.tp_world == 1
        while True:
            with self._work:
                while (
                    self._running
                    and not self._pending
                    and not self.scheduler.has_work()
                    and pending is None
                ):
                    self._work.wait(timeout=0.2)
                if not self._running:
                    break
                new, self._pending = self._pending, []
                cancels, self._cancelled = self._cancelled, []
            for req in new:
                self.scheduler.add_request(req)
            for req in cancels:
                self.scheduler.cancel_request(req)
            try:
                t0 = time.monotonic()
                if pending is None:
                    pending = self._schedule_and_launch(None)
                    self._m["busy_time_s"] += time.monotonic() - t0
                    continue
                # a step is in flight: launch its successor speculatively,
                # then sync/commit the pending one.  A detected forced run
                # requests ONE spec bail so the following commit can fold
                # it into a prefill chunk (no in-flight successor to
                # invalidate).
                do_spec = spec_ok and not self._spec_bail
                self._spec_bail = False
                nxt = self._schedule_and_launch(pending["batch"]) if do_spec else None
                if nxt is not None:
                    self._m["spec_steps"] = self._m.get("spec_steps", 0) + 1
                t2 = time.monotonic()
                self._commit(pending, has_successor=nxt is not None)
                self._m["sample_time_s"] += time.monotonic() - t2
                if nxt is not None and nxt.get("deferred"):
                    # grammar-carrying speculative step: its forward is
                    # already on the GPU; launch the sampler now that the
                    # commit above advanced the grammar states its masks
                    # depend on
                    t3 = time.monotonic()
                    nxt = self._launch_sample(nxt["batch"], nxt["logits"])
                    self._m["sample_launch_time_s"] = (
                        self._m.get("sample_launch_time_s", 0.0) + time.monotonic() - t3
                    )
                pending = nxt
                self._m["busy_time_s"] += time.monotonic() - t0
            except Exception as e:  # noqa: BLE001 — fail all in-flight requests
                import traceback

                traceback.print_exc()
                pending = None
                for s in list(self.scheduler.running) + list(self.scheduler.waiting):
                    self.scheduler.abort(s, e)
        # drain on shutdown so waiters are not left hanging
        if pending is not None:
            try:
                if pending.get("deferred"):
                    pending = self._launch_sample(pending["batch"], pending["logits"])
                self._commit(pending)
            except Exception:
                pass

    def _schedule_and_launch(self, spec_after):
        """schedule (optionally speculative) + forward + sample launch;
        returns the pending-step dict or None."""
        t0 = time.monotonic()
        out = self.scheduler.schedule(spec_after=spec_after)
        t1 = time.monotonic()
        self._m["preemptions"] += len(out.preempted)
        self._m["sched_time_s"] += t1 - t0
        if out.batch is None:
            return None
        batch = out.batch
        if batch.spec_src_rows is not None and spec_after is not None:
            # decode-row tokens: gather from the previous step's GPU samples
            P = batch.num_prefill_tokens
            batch.token_ids[P:] = spec_after._tokens_gpu[batch.spec_src_rows]
        if self.tp_world > 1 and self.tp_rank == 0:
            from ..engine.batch import batch_to_wire
            from ..parallel.tp import broadcast_step

            broadcast_step(batch_to_wire(batch))
        logits = None
        if self.graph_runner is not None and not batch.prefills:
            logits = self.graph_runner.run(batch)
            if logits is not None:
                self._m["graph_steps"] = self._m.get("graph_steps", 0) + 1
        if logits is None:
            logits = self.model.forward(batch)
        self._m["compute_time_s"] += time.monotonic() - t1
        self._m["steps"] += 1
        self._m["prompt_tokens"] += batch.num_prefill_tokens
        if spec_after is not None and any(
            self._stateful_sampling(sid) for sid in batch.sample_seq_ids
        ):
            # grammar masks, repetition penalties and per-request seeded
            # streams all depend on the COMMITTED output: defer the sampler
            # launch to after the predecessor's commit (the loop resolves
            # this before scheduling OUR successor, so _tokens_gpu exists
            # by the time it is gathered from)
            return {"batch": batch, "logits": logits, "deferred": True}
        t2 = time.monotonic()
        pending = self._launch_sample(batch, logits)
        self._m["sample_launch_time_s"] = (
            self._m.get("sample_launch_time_s", 0.0) + time.monotonic() - t2
        )
        return pending

    def step(self) -> bool:
        """One serial scheduler round + forward + sampling (tests and the
        TP worker path).  Returns True if any work ran."""
        pending = self._schedule_and_launch(None)
        if pending is None:
            return False
        self._commit(pending)
        return True

    def _stateful_sampling(self, sid) -> bool:
        s = self.scheduler.seq_by_id(sid)
        if s is None:
            return False
        if s.grammar is not None:
            return True
        sp = s.request.sampling
        return (
            sp.seed is not None
            or bool(sp.frequency_penalty)
            or bool(sp.presence_penalty)
        )

    def _launch_sample(self, batch, logits: torch.Tensor):
        """Build sampling inputs and launch the sampler (async on GPU);
        returns the pending-step record for a later _commit."""
        ta = time.monotonic()
        seqs: List[Sequence] = []
        for sid in batch.sample_seq_ids:
            seqs.append(self.scheduler.seq_by_id(sid))
        if not seqs:
            return {"batch": batch, "seqs": [], "tokens": None}
        B = len(seqs)
        live = logits[:, :self._live]  # sampling restricted to decodable ids
        self._m["ls_gather_s"] = self._m.get("ls_gather_s", 0.0) + time.monotonic() - ta
        tb = time.monotonic()
        # per-sequence sampling params are static: refill the persistent
        # staging buffers only when the batch's id tuple changes (a row's
        # sequence may have finished at the commit that just ran — sample
        # it with defaults, the commit discards its token)
        key = tuple(batch.sample_seq_ids)
        if self._sampling_cache_key != key:
            ht, hk, hp = self._ps_temp.host(), self._ps_topk.host(), self._ps_topp.host()
            for i, s in enumerate(seqs):
                p = s.request.sampling if s is not None else None
                ht[i] = p.temperature if p else 0.0
                hk[i] = p.top_k if p else 0
                hp[i] = p.top_p if p else 1.0
            self._sampling_cache_key = key
            self._sampling_cache = (
                self._ps_temp.commit(B),
                self._ps_topk.commit(B),
                self._ps_topp.commit(B),
            )
        temps, top_ks, top_ps = self._sampling_cache
        mask = None
        mask_map = None
        if any(s is not None and s.grammar is not None for s in seqs):
            tm = time.monotonic()
            if self._ps_maskmap is None:
                # dense path (byte vocab)
                m = self._ps_mask.host()
                m[:B] = True
                for i, s in enumerate(seqs):
                    if s is not None and s.grammar is not None:
                        allowed = s.grammar.allowed_tokens()
                        if s.grammar.accepting:
                            allowed = set(allowed) | {self._eot}
                        row = m[i]
                        row[:] = False
                        row[list(allowed)] = True
                mask = self._ps_mask.commit(B)
            else:
                # compact path (BPE vocab): one mask row per constrained seq
                m = self._ps_mask.host()
                mm = self._ps_maskmap.host()
                mm[:B] = -1
                j = 0
                overflow = []
                for i, s in enumerate(seqs):
                    if s is None or s.grammar is None:
                        continue
                    allowed = s.grammar.allowed_tokens()
                    if s.grammar.accepting:
                        allowed = set(allowed) | {self._eot}
                    if j < self._mask_rows:
                        row = m[j]
                        row[:] = False
                        import numpy as _np2

                        row[_np2.fromiter(allowed, dtype=_np2.int64, count=len(allowed))] = True
                        mm[i] = j
                        j += 1
                    else:
                        overflow.append((i, allowed))
                mask = self._ps_mask.commit(max(1, j))
                mask_map = self._ps_maskmap.commit(B)
                for i, allowed in overflow:
                    # slow path beyond the compact-row cap: mask in place
                    ids = torch.tensor(sorted(allowed), dtype=torch.long,
                                       device=live.device)
                    rowmask = torch.zeros(self._live, dtype=torch.bool,
                                          device=live.device)
                    rowmask[ids] = True
                    live[i] = live[i].masked_fill(~rowmask, float("-inf"))
            self._m["mask_time_s"] = (
                self._m.get("mask_time_s", 0.0) + time.monotonic() - tm
            )
        self._m["ls_params_s"] = self._m.get("ls_params_s", 0.0) + time.monotonic() - tb
        tc = time.monotonic()
        # frequency/presence penalties (llm_types.go BaseConfig): subtract
        # freq·count + presence·[seen] over the request's committed output
        if any(
            s is not None and s.state != "finished"
            and (s.request.sampling.frequency_penalty or s.request.sampling.presence_penalty)
            for s in seqs
        ):
            import numpy as np

            if self._live <= 512:
                pen = np.zeros((B, self._live), dtype=np.float32)
                for i, s in enumerate(seqs):
                    if s is None or s.state == "finished":
                        continue
                    sp2 = s.request.sampling
                    if not (sp2.frequency_penalty or sp2.presence_penalty):
                        continue
                    counts: Dict[int, int] = {}
                    for t in s.request.output_ids:
                        if t < self._live:
                            counts[t] = counts.get(t, 0) + 1
                    for t, c in counts.items():
                        pen[i, t] = sp2.frequency_penalty * c + sp2.presence_penalty
                live = live - self.scheduler.stager.tensor("pen", pen, "float32")
            else:
                # BPE vocab: sparse in-place scatter (a dense [B, 128k]
                # penalty tensor would be hundreds of MB per step)
                rows, cols, vals = [], [], []
                for i, s in enumerate(seqs):
                    if s is None or s.state == "finished":
                        continue
                    sp2 = s.request.sampling
                    if not (sp2.frequency_penalty or sp2.presence_penalty):
                        continue
                    counts = {}
                    for t in s.request.output_ids:
                        if t < self._live:
                            counts[t] = counts.get(t, 0) + 1
                    for t, c in counts.items():
                        rows.append(i)
                        cols.append(t)
                        vals.append(sp2.frequency_penalty * c + sp2.presence_penalty)
                if rows:
                    ri = torch.tensor(rows, dtype=torch.long, device=live.device)
                    ci = torch.tensor(cols, dtype=torch.long, device=live.device)
                    vi = torch.tensor(vals, dtype=torch.float32, device=live.device)
                    live = live.clone() if live._base is not None else live
                    live[ri, ci] -= vi.to(live.dtype)
        # per-request seeded draws: each seeded request consumes exactly one
        # uniform from its own generator per committed token
        uniforms = None
        if any(
            s is not None and s.state != "finished"
            and s.request.sampling.seed is not None
            for s in seqs
        ):
            uniforms = torch.rand(B, device=logits.device, generator=self._gen)
            for i, s in enumerate(seqs):
                if s is None or s.state == "finished" or s.request.sampling.seed is None:
                    continue
                if s.request.gen is None:
                    g = torch.Generator(device=logits.device)
                    g.manual_seed(int(s.request.sampling.seed))
                    s.request.gen = g
                uniforms[i] = torch.rand(1, device=logits.device, generator=s.request.gen)[0]
        tokens = ops.softmax_sample(
            live, temps, top_ks, top_ps, self._gen, mask, uniforms, mask_map
        )
        self._m["ls_kernel_s"] = self._m.get("ls_kernel_s", 0.0) + time.monotonic() - tc
        batch._tokens_gpu = tokens  # speculative successors gather from this
        return {"batch": batch, "seqs": seqs, "tokens": tokens}

    def _commit(self, pending, has_successor: bool = False) -> None:
        """Sync the sampled tokens and advance sequence/grammar state."""
        seqs = pending["seqs"]
        if not seqs:
            return
        tokens_cpu = pending["tokens"].tolist()  # the step's one sync point
        for s, tok in zip(seqs, tokens_cpu):
            if s is None or s.state == "finished":
                continue  # finished at an earlier commit (speculative row)
            self.scheduler.append_sampled(s, tok)
            self._m["generated_tokens"] += 1
            if s.request.first_token_time is None:
                s.request.first_token_time = time.monotonic()
            if s.grammar is not None:
                if tok == self._eot and s.grammar.accepting:
                    s.grammar.advance(self._eot)
                    s.request.output_ids.pop()  # EOT is not part of the JSON
                    s.output_ids.pop()
                    self._m["requests_completed"] += 1
                    self.scheduler.finish_seq(s, "tool_calls")
                    continue
                s.grammar.advance(tok)
                if len(s.request.output_ids) >= s.request.sampling.max_tokens * 8 or (
                    s.total_len >= self._ctx_limit
                ):
                    # runaway guard + hard context-limit stop: positions must
                    # never walk past the RoPE table (cfg.max_position)
                    self._m["requests_completed"] += 1
                    self.scheduler.finish_seq(s, "length")
                    continue
                # forced-run folding: a run of singleton-allowed bytes (MID
                # scaffolding, schema literals, a lone tool name's tail)
                # becomes one chunked-prefill extension instead of one
                # masked decode step per byte
                if self.cfg.grammar_fold and s.state == "decode":
                    forced = s.grammar.forced_run()
                    if len(forced) >= 4 and (
                        s.total_len + len(forced) < self._ctx_limit - 1
                    ):
                        if has_successor:
                            # the in-flight speculative step would sample a
                            # stale position after a fold: bail spec once so
                            # the NEXT commit folds with nothing in flight
                            self._spec_bail = True
                        else:
                            self._fold_forced(s, forced)
                continue
            if tok == self._eot:
                s.request.output_ids.pop()
                s.output_ids.pop()
                self._m["requests_completed"] += 1
                self.scheduler.finish_seq(s, "stop")
                continue
            if s.request.on_token is not None:
                s.request.on_token(tok)
            if len(s.request.output_ids) >= s.request.sampling.max_tokens or (
                s.total_len >= self._ctx_limit
            ):
                # request-level count: survives recompute preemption (the
                # per-sequence list folds into the prompt on preempt)
                self._m["requests_completed"] += 1
                self.scheduler.finish_seq(s, "length")

    def _fold_forced(self, s, forced) -> None:
        """Fold a forced byte run into the sequence's prompt so the next
        step prefills it as ONE chunk (with its following logits row)
        instead of len(forced) sequential masked decode steps.

        Bookkeeping mirrors recompute-preemption's prompt fold
        (scheduler._preempt) minus the KV free: the KV-resident prefix
        stays (= bm.seq_len: every token except the just-sampled one),
        the folded bytes count as generated output, and the grammar
        advances through them now so the next sampled token sees the
        post-scaffolding state."""
        for b in forced:
            s.request.output_ids.append(b)
            s.grammar.advance(b)
            self._m["generated_tokens"] += 1
        s.prompt_ids = s.prompt_ids + s.output_ids + list(forced)
        s.output_ids = []
        s.num_processed = self.bm.seq_len(s.seq_id)
        s.state = "prefill"
        self._m["grammar_folds"] = self._m.get("grammar_folds", 0) + 1
        self._m["grammar_folded_tokens"] = (
            self._m.get("grammar_folded_tokens", 0) + len(forced)
        )

    # ------------------------------------------------------------- metrics

    def metrics(self) -> Dict[str, float]:
        m = dict(self._m)
        elapsed = time.monotonic() - self._start_time
        m["uptime_s"] = elapsed
        m["tokens_per_s"] = self._m["generated_tokens"] / max(1e-9, elapsed)
        m["kv_occupancy"] = self.bm.used_blocks / max(1, self.bm.num_blocks)
        m["running_seqs"] = len(self.scheduler.running)
        m["waiting_seqs"] = len(self.scheduler.waiting)
        m["retired_seqs"] = len(self.scheduler.retired)
        m["continuation_hits"] = self.scheduler.continuation_hits
        m["continuation_tokens_saved"] = self.scheduler.continuation_tokens_saved
        return m
