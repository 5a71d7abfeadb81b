"""Inference request/response types."""
from __future__ import annotations

import dataclasses
import threading
import time
from typing import Any, Dict, List, Optional


@dataclasses.dataclass
class SamplingParams:
    max_tokens: int = 256
    temperature: float = 0.7
    top_p: float = 1.0
    top_k: int = 0
    frequency_penalty: float = 0.0
    presence_penalty: float = 0.0
    seed: Optional[int] = None
    # "auto" | "required" | "none" — whether the turn must produce a tool call
    # (grammar-constrained JSON).  With random-init weights "auto" resolves by
    # the workload policy in engine.chat(); a trained checkpoint would resolve
    # it from the model's own TOOL_CALL_START logit.
    tool_choice: str = "auto"


@dataclasses.dataclass
class ChatResult:
    text: str = ""
    tool_calls: List[Dict[str, Any]] = dataclasses.field(default_factory=list)
    prompt_tokens: int = 0
    completion_tokens: int = 0
    finish_reason: str = "stop"   # stop | length | tool_calls
    latency_s: float = 0.0


class InferenceRequest:
    """One sequence through the engine; completed via a threading future."""

    _counter = [0]
    _counter_lock = threading.Lock()

    def __init__(self, prompt_ids: List[int], sampling: SamplingParams,
                 constrained: bool = False, tools: Optional[List[Dict]] = None,
                 pre_in_prompt: bool = False):
        with self._counter_lock:
            self._counter[0] += 1
            self.request_id = self._counter[0]
        self.prompt_ids = prompt_ids
        self.sampling = sampling
        self.constrained = constrained
        # True when the caller appended ToolCallGrammar.PRE ('{"name": "')
        # to the prompt: the forced preamble then costs one prefill chunk
        # instead of len(PRE) sequential masked decode steps
        self.pre_in_prompt = pre_in_prompt
        self.tools = tools or []
        self.output_ids: List[int] = []
        self.submit_time = time.monotonic()
        self.first_token_time: Optional[float] = None
        self.finish_reason = "stop"
        self._event = threading.Event()
        self._error: Optional[BaseException] = None
        # invoked (from the engine thread) when the request completes; used
        # by the async serving path to requeue the owning Task reconcile
        self.on_complete = None
        # invoked (from the engine thread) per committed free-text token —
        # the streaming path; must be cheap and non-blocking (queue.put)
        self.on_token = None
        # per-request RNG for SamplingParams.seed (created on first use by
        # the engine, on its device)
        self.gen = None
        # scheduler state
        self.seq = None  # assigned by the scheduler

    def finish(self, reason: str) -> None:
        self.finish_reason = reason
        self._event.set()
        if self.on_complete is not None:
            self.on_complete(self)

    def fail(self, err: BaseException) -> None:
        self._error = err
        self._event.set()
        if self.on_complete is not None:
            self.on_complete(self)

    @property
    def error(self) -> Optional[BaseException]:
        return self._error

    def wait(self, timeout: Optional[float] = None) -> List[int]:
        if not self._event.wait(timeout):
            raise TimeoutError(f"request {self.request_id} timed out")
        if self._error is not None:
            raise self._error
        return self.output_ids
