"""Task reconciler — the agent loop.

Parity with acp/internal/controller/task/ (task_controller.go + state_machine.go,
1,435 LoC): phase-dispatched state machine over

  "" → Initializing → validate(Pending) → ReadyForLLM → [LLM turn]
      → FinalAnswer | ToolCallsPending → (tool join) → ReadyForLLM → …

with the context window checkpointed to the store on every transition
(status.contextWindow is the source of truth and the crash-resume point,
task_types.go:139, state_machine.go:665-669), dual-layer locking around the
LLM call (in-memory per-task mutex + store lease "task-llm-<name>",
state_machine.go:944-966, 1069-1145), 4xx-terminal LLM error handling
(733-789), parallel ToolCall fan-out with deterministic child names
"<task>-<reqid>-tc-%02d" (676-731), and the v1beta3 respond_to_human final
answer path (968-1067).

Differences by design (MI355X-first):
- the tool-call join is watch-driven: this controller ``owns`` ToolCalls, so
  a child status flip requeues the parent immediately instead of the
  reference's 5 s poll (the dominant latency term in BASELINE.md);
- the LLM call runs against the in-process engine for provider=local — the
  engine continuously batches every concurrent task's request into one
  paged-KV scheduler, so reconciler workers block only on their own turn's
  completion future.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Dict, List, Optional, Tuple

from ..api.types import (
    AGENT,
    CONTACT_CHANNEL,
    LLM,
    SECRET,
    TASK,
    TOOL_CALL,
    Message,
    TaskPhase,
    TaskStatusType,
    ToolType,
    owner_ref,
)
from ..api.validation import (
    RetryableValidationError,
    ValidationError,
    generate_k8s_random_string,
    get_user_message_preview,
    validate_contact_channel_ref,
    validate_task_message_input,
)
from ..llmclient.base import LLMRequestError, Tool, ToolFunction, tool_from_contact_channel
from ..llmclient.factory import LLMClientFactory
from ..mcp.adapter import convert_mcp_tools_to_llm_tools
from ..tracing import get_tracer, reconstruct_span_context
from .manager import Reconciler, Result

DEFAULT_REQUEUE = 5.0           # task_controller.go:23
LEASE_DURATION = 30.0           # state_machine.go:80
LEASE_RETRY = 5.0               # state_machine.go:179
V1BETA3_LABEL = "acp.humanlayer.dev/v1beta3"
LABEL_TASK = "acp.humanlayer.dev/task"
LABEL_TCREQ = "acp.humanlayer.dev/toolcallrequest"


class TaskReconciler(Reconciler):
    kind = TASK
    owns = (TOOL_CALL,)
    workers = 8

    def __init__(
        self,
        store,
        llm_client_factory: Optional[LLMClientFactory] = None,
        mcp_manager=None,
        humanlayer_factory=None,
        pod_name: str = "acp-controller-0",
    ):
        super().__init__(store)
        self.factory = llm_client_factory or LLMClientFactory()
        self.mcp = mcp_manager
        self.humanlayer = humanlayer_factory
        self.pod_name = pod_name
        self.tracer = get_tracer()
        self._mutexes: Dict[Tuple[str, str], threading.Lock] = {}
        self._mutex_guard = threading.Lock()
        # in-flight async LLM turns, keyed by (ns, name): workers submit to
        # the engine and return; the completion callback requeues the task
        # (SURVEY.md §7 step 5 — no one-blocking-call-per-reconcile)
        self._inflight: Dict[Tuple[str, str], dict] = {}
        self._manager = None  # back-ref set by ControllerManager.register

    def on_deleted(self, name: str, namespace: str) -> None:
        """Drop per-task state when the Task object goes away — the mutex
        and in-flight maps otherwise grow one entry per task forever under
        create/delete churn."""
        key = (namespace, name)
        self._inflight.pop(key, None)
        with self._mutex_guard:
            self._mutexes.pop(key, None)

    # ToolCall events requeue the parent task via the task label
    def map_owned(self, ev):
        labels = ev.obj.get("metadata", {}).get("labels", {}) or {}
        task = labels.get(LABEL_TASK)
        if task:
            return task, ev.obj["metadata"].get("namespace", "default")
        return super().map_owned(ev)

    def _task_mutex(self, name: str, namespace: str) -> threading.Lock:
        """In-memory per-task mutex (state_machine.go:38-44, 944-966)."""
        with self._mutex_guard:
            key = (namespace, name)
            if key not in self._mutexes:
                self._mutexes[key] = threading.Lock()
            return self._mutexes[key]

    # ------------------------------------------------------------- dispatch

    def reconcile(self, name: str, namespace: str) -> Result:
        task = self.store.get(TASK, name, namespace)
        if task is None:
            return Result()
        phase = task.get("status", {}).get("phase", "")
        if phase == "":
            return self._initialize(task)
        if phase in (TaskPhase.INITIALIZING, TaskPhase.PENDING):
            return self._validate(task)
        if phase == TaskPhase.READY_FOR_LLM:
            return self._send_llm_request(task)
        if phase == TaskPhase.TOOL_CALLS_PENDING:
            return self._check_tool_calls(task)
        if phase in (TaskPhase.FINAL_ANSWER, TaskPhase.FAILED):
            return self._handle_terminal(task)
        if phase == TaskPhase.ERROR_BACKOFF:
            task["status"]["phase"] = TaskPhase.READY_FOR_LLM
            self.store.update_status(task)
            return Result(requeue=True)
        return Result()

    # ----------------------------------------------------------- initialize

    def _initialize(self, task) -> Result:
        """Create the root span and persist its ids (state_machine.go:119-145)."""
        root = self.tracer.start("Task", attributes={"task.name": task["metadata"]["name"]})
        status = task.setdefault("status", {})
        status.update(
            {
                "phase": TaskPhase.INITIALIZING,
                "status": TaskStatusType.PENDING,
                "statusDetail": "Initializing Task",
                "spanContext": {"traceID": root.trace_id, "spanID": root.span_id},
                "startTime": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            }
        )
        self.store.update_status(task)
        return Result(requeue=True)

    # ------------------------------------------------------------- validate

    def _validate(self, task) -> Result:
        """validateTaskAndAgent + prepareForLLM (state_machine.go:379-463)."""
        status = task["status"]
        spec = task.get("spec", {})
        ns = task["metadata"].get("namespace", "default")
        agent_name = (spec.get("agentRef") or {}).get("name", "")
        agent = self.store.get(AGENT, agent_name, ns)
        if agent is None:
            status.update(
                {
                    "phase": TaskPhase.PENDING,
                    "status": TaskStatusType.PENDING,
                    "statusDetail": f'Waiting for Agent "{agent_name}" to exist',
                }
            )
            self.store.record_event(task, "Normal", "Waiting", "Waiting for Agent to exist")
            self.store.update_status(task)
            return Result(requeue_after=DEFAULT_REQUEUE)
        if not agent.get("status", {}).get("ready", False):
            status.update(
                {
                    "phase": TaskPhase.PENDING,
                    "status": TaskStatusType.PENDING,
                    "statusDetail": f'Waiting for agent "{agent_name}" to become ready',
                }
            )
            self.store.record_event(
                task, "Normal", "Waiting", f'Waiting for agent "{agent_name}" to become ready'
            )
            self.store.update_status(task)
            return Result(requeue_after=DEFAULT_REQUEUE)

        # prepareForLLM (426-463)
        try:
            validate_task_message_input(
                spec.get("userMessage", ""), spec.get("contextWindow", [])
            )
            validate_contact_channel_ref(self.store, task)
        except RetryableValidationError as e:
            status.update(
                {
                    "phase": TaskPhase.PENDING,
                    "status": TaskStatusType.PENDING,
                    "statusDetail": str(e),
                }
            )
            self.store.record_event(task, "Normal", "Waiting", str(e))
            self.store.update_status(task)
            return Result(requeue_after=DEFAULT_REQUEUE)
        except ValidationError as e:
            status.update(
                {
                    "phase": TaskPhase.FAILED,
                    "ready": False,
                    "status": TaskStatusType.ERROR,
                    "statusDetail": str(e),
                    "error": str(e),
                }
            )
            self.store.record_event(task, "Warning", "ValidationFailed", str(e))
            self.store.update_status(task)
            return Result()

        context_window = self._build_initial_context_window(task, agent)
        status.update(
            {
                "phase": TaskPhase.READY_FOR_LLM,
                "ready": True,
                "status": TaskStatusType.READY,
                "statusDetail": "Ready to send to LLM",
                "contextWindow": context_window,
                "messageCount": len(context_window),
                "userMsgPreview": get_user_message_preview(
                    spec.get("userMessage", ""), spec.get("contextWindow", [])
                ),
                "error": "",
            }
        )
        self.store.record_event(task, "Normal", "ValidationSucceeded", "Task validation succeeded")
        self.store.update_status(task)
        return Result(requeue=True)

    @staticmethod
    def _build_initial_context_window(task, agent) -> List[dict]:
        """task_helpers.go:13-44: system + user, or spec.contextWindow with the
        agent's system prompt ensured at position 0."""
        spec = task.get("spec", {})
        system = agent.get("spec", {}).get("system", "")
        cw = [dict(m) for m in spec.get("contextWindow", []) or []]
        if cw:
            if not any(m.get("role") == "system" for m in cw):
                cw.insert(0, {"role": "system", "content": system})
            return cw
        return [
            {"role": "system", "content": system},
            {"role": "user", "content": spec.get("userMessage", "")},
        ]

    # -------------------------------------------------------------- LLM turn

    def _collect_tools(self, task, agent) -> List[Tool]:
        """MCP tools + contact-channel tools + delegate-to-agent tools
        (state_machine.go:540-583, task_controller.go:94-117)."""
        ns = task["metadata"].get("namespace", "default")
        tools: List[Tool] = []
        for ref in agent.get("spec", {}).get("mcpServers", []) or []:
            mcp_tools = self.mcp.get_tools(ref["name"]) if self.mcp else None
            if mcp_tools is None:
                srv = self.store.get("MCPServer", ref["name"], ns)
                mcp_tools = (srv or {}).get("status", {}).get("tools", []) or []
            tools.extend(convert_mcp_tools_to_llm_tools(mcp_tools, ref["name"]))
        for ref in agent.get("spec", {}).get("humanContactChannels", []) or []:
            ch = self.store.get(CONTACT_CHANNEL, ref["name"], ns)
            if ch is not None:
                tools.append(tool_from_contact_channel(ch))
        for ref in agent.get("spec", {}).get("subAgents", []) or []:
            sub = self.store.get(AGENT, ref["name"], ns)
            desc = (sub or {}).get("spec", {}).get("description", "") or (
                f'Delegate a task to agent "{ref["name"]}"'
            )
            tools.append(
                Tool(
                    function=ToolFunction(
                        name=f'delegate_to_agent__{ref["name"]}',
                        description=desc,
                        parameters={
                            "type": "object",
                            "properties": {"message": {"type": "string"}},
                            "required": ["message"],
                        },
                    ),
                    acp_tool_type=ToolType.DELEGATE_TO_AGENT,
                )
            )
        return tools

    def _send_llm_request(self, task) -> Result:
        """state_machine.go:162-288 — the hot loop."""
        name = task["metadata"]["name"]
        ns = task["metadata"].get("namespace", "default")
        mutex = self._task_mutex(name, ns)
        if not mutex.acquire(blocking=False):
            return Result(requeue_after=0.05)
        try:
            lease_name = f"task-llm-{name}"
            if not self.store.acquire_lease(lease_name, self.pod_name, LEASE_DURATION, ns):
                return Result(requeue_after=LEASE_RETRY)
            try:
                return self._send_llm_request_locked(task)
            finally:
                self.store.release_lease(lease_name, self.pod_name, ns)
        finally:
            mutex.release()

    def _send_llm_request_locked(self, task) -> Result:
        # re-read under the lock: another pod may have advanced the phase
        name = task["metadata"]["name"]
        ns = task["metadata"].get("namespace", "default")
        fresh = self.store.get(TASK, name, ns)
        if fresh is None or fresh["status"].get("phase") != TaskPhase.READY_FOR_LLM:
            self._inflight.pop((ns, name), None)
            return Result()
        task = fresh
        status = task["status"]
        spec = task.get("spec", {})
        key = (ns, name)
        labels = task["metadata"].get("labels", {}) or {}
        if labels.get(V1BETA3_LABEL) == "true" and self._v1beta3_parked(status):
            if status.get("statusDetail") != "Awaiting next inbound event":
                status["statusDetail"] = "Awaiting next inbound event"
                self.store.update_status(task)
            return Result()  # parked: a new inbound event opens a new Task
        marker = len(status.get("contextWindow", []))

        # a completed async turn? process it under the lock/lease
        entry = self._inflight.get(key)
        if entry is not None:
            if entry["marker"] != marker:
                # context moved on while the turn was in flight — drop it
                self._inflight.pop(key, None)
            elif not entry["done"]:
                return Result()  # completion callback will requeue
            else:
                self._inflight.pop(key, None)
                if entry["error"] is not None:
                    return self._handle_llm_error(task, entry["error"])
                return self._process_llm_response(task, entry["output"], entry["tools"])

        agent = self.store.get(AGENT, (spec.get("agentRef") or {}).get("name", ""), ns)
        if agent is None:
            return Result(requeue_after=DEFAULT_REQUEUE)

        # LLM + credentials (480-538)
        llm_name = (agent.get("spec", {}).get("llmRef") or {}).get("name", "")
        llm = self.store.get(LLM, llm_name, ns)
        if llm is None:
            self.store.record_event(task, "Warning", "LLMFetchFailed", f'LLM "{llm_name}" not found')
            return Result(requeue_after=DEFAULT_REQUEUE)
        api_key = ""
        src = (llm.get("spec", {}).get("apiKeyFrom") or {}).get("secretKeyRef", {})
        if src.get("name"):
            secret = self.store.get(SECRET, src["name"], ns)
            if secret is None:
                self.store.record_event(
                    task, "Warning", "APIKeySecretFetchFailed", f'secret "{src["name"]}" not found'
                )
                return Result(requeue_after=DEFAULT_REQUEUE)
            data = secret.get("spec", {}).get("data", {}) or secret.get("data", {})
            api_key = str(data.get(src.get("key", ""), ""))

        tools = self._collect_tools(task, agent)
        messages = [Message.from_dict(m) for m in status.get("contextWindow", [])]

        self.store.record_event(
            task, "Normal", "SendingContextWindowToLLM", "Sending context window to LLM"
        )

        # child LLMRequest span carrying message/tool counts (585-603)
        parent = None
        sc = status.get("spanContext") or {}
        if sc.get("traceID"):
            try:
                parent = reconstruct_span_context(sc["traceID"], sc["spanID"])
            except ValueError:
                parent = None
        llm_span = self.tracer.start(
            "LLMRequest",
            parent=parent,
            attributes={"messages": len(messages), "tools": len(tools)},
        )
        try:
            client = self.factory.create_client(llm, api_key)
        except Exception as e:
            llm_span.record_error(e)
            llm_span.set_status("ERROR", str(e))
            llm_span.end()
            return self._handle_llm_error(task, e)

        entry = {
            "marker": marker,
            "done": False,
            "output": None,
            "error": None,
            "tools": tools,
            "span": llm_span,
        }
        self._inflight[key] = entry

        def _on_turn(output, error, _entry=entry, _key=key):
            span = _entry["span"]
            if error is not None:
                span.record_error(error)
                span.set_status("ERROR", str(error))
            else:
                span.set_status("OK")
            span.end()
            _entry["output"] = output
            _entry["error"] = error
            _entry["done"] = True
            if self._manager is not None:
                self._manager.enqueue(TASK, _key[1], _key[0])

        client.send_request_async(messages, tools, _on_turn)

        # synchronous clients (mock, remote stub) complete inline — finish
        # the turn in this reconcile instead of an extra queue hop
        if entry["done"] and self._inflight.get(key) is entry:
            self._inflight.pop(key, None)
            if entry["error"] is not None:
                return self._handle_llm_error(task, entry["error"])
            return self._process_llm_response(task, entry["output"], entry["tools"])
        return Result()

    def _handle_llm_error(self, task, err) -> Result:
        """4xx terminal vs retry-in-place (state_machine.go:733-789)."""
        status = task["status"]
        is_4xx = isinstance(err, LLMRequestError) and 400 <= err.status_code < 500
        if is_4xx:
            status.update(
                {
                    "ready": False,
                    "status": TaskStatusType.ERROR,
                    "phase": TaskPhase.FAILED,
                    "statusDetail": f"LLM request failed: {err}",
                    "error": str(err),
                }
            )
            self.store.record_event(task, "Warning", "LLMRequestFailed4xx", str(err))
            self.store.update_status(task)
            return Result()
        status.update(
            {
                "ready": False,
                "status": TaskStatusType.ERROR,
                "statusDetail": f"LLM request failed: {err}",
                "error": str(err),
            }
        )
        self.store.record_event(task, "Warning", "LLMRequestFailed", str(err))
        self.store.update_status(task)
        return Result(requeue_after=DEFAULT_REQUEUE)

    def _process_llm_response(self, task, output: Message, tools: List[Tool]) -> Result:
        """state_machine.go:605-674."""
        status = task["status"]
        labels = task["metadata"].get("labels", {}) or {}
        prev_phase = status.get("phase", "")
        if output.content:
            if labels.get(V1BETA3_LABEL) == "true":
                # reference parity: EVERY content output on a v1beta3 task
                # becomes a respond_to_human ToolCall (state_machine.go:
                # 609-611 has no round-done special case); the loop back to
                # ReadyForLLM then parks (see _send_llm_request_locked)
                return self._v1beta3_final_answer(task, output)
            status.update(
                {
                    "output": output.content,
                    "phase": TaskPhase.FINAL_ANSWER,
                    "ready": True,
                    "status": TaskStatusType.READY,
                    "statusDetail": "LLM final response received",
                    "error": "",
                    "completionTime": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
                }
            )
            status.setdefault("contextWindow", []).append(
                {"role": "assistant", "content": output.content}
            )
            status["messageCount"] = len(status["contextWindow"])
            if prev_phase != TaskPhase.FINAL_ANSWER:
                self.store.record_event(
                    task, "Normal", "LLMFinalAnswer", "LLM response received successfully"
                )
            self.store.update_status(task)
            # async HumanLayer notification when a contact channel is set (841-860)
            if task.get("spec", {}).get("contactChannelRef") and self.humanlayer is not None:
                threading.Thread(
                    target=self._notify_final_result,
                    args=(task, output.content),
                    daemon=True,
                ).start()
            return Result()

        # tool-call branch
        req_id = generate_k8s_random_string(7)
        status.update(
            {
                "output": "",
                "phase": TaskPhase.TOOL_CALLS_PENDING,
                "toolCallRequestId": req_id,
                "ready": True,
                "status": TaskStatusType.READY,
                "statusDetail": "LLM response received, tool calls pending",
                "error": "",
            }
        )
        status.setdefault("contextWindow", []).append(
            {"role": "assistant", "content": "", "toolCalls": [tc.to_dict() for tc in output.tool_calls]}
        )
        status["messageCount"] = len(status["contextWindow"])
        self.store.record_event(
            task, "Normal", "ToolCallsPending", "LLM response received, tool calls pending"
        )
        self.store.update_status(task)
        return self._create_tool_calls(task, output.tool_calls, tools)

    def _create_tool_calls(self, task, tool_calls, tools: List[Tool]) -> Result:
        """Deterministic child names + labels + ownerRef (state_machine.go:676-731)."""
        name = task["metadata"]["name"]
        ns = task["metadata"].get("namespace", "default")
        req_id = task["status"]["toolCallRequestId"]
        tool_type_map = {t.function.name: t.acp_tool_type for t in tools}
        for i, tc in enumerate(tool_calls):
            new_name = f"{name}-{req_id}-tc-{i + 1:02d}"
            obj = {
                "apiVersion": "acp.humanlayer.dev/v1alpha1",
                "kind": TOOL_CALL,
                "metadata": {
                    "name": new_name,
                    "namespace": ns,
                    "labels": {LABEL_TASK: name, LABEL_TCREQ: req_id},
                    "ownerReferences": [owner_ref(task)],
                },
                "spec": {
                    "toolCallId": tc.id,
                    "taskRef": {"name": name},
                    "toolRef": {"name": tc.function.name},
                    "toolType": tool_type_map.get(tc.function.name, ""),
                    "arguments": tc.function.arguments,
                },
                "status": {},
            }
            try:
                self.store.create(obj)
            except Exception:
                continue  # idempotent: already created by a previous attempt
            self.store.record_event(task, "Normal", "ToolCallCreated", f"Created ToolCall {new_name}")
        # watch-driven join: ToolCall status flips requeue us; the delay is a
        # crash-safety net, not the join mechanism
        return Result(requeue_after=DEFAULT_REQUEUE)

    # ----------------------------------------------------------- tool join

    def _check_tool_calls(self, task) -> Result:
        """state_machine.go:291-341."""
        name = task["metadata"]["name"]
        ns = task["metadata"].get("namespace", "default")
        status = task["status"]
        req_id = status.get("toolCallRequestId", "")
        tool_calls = self.store.list(
            TOOL_CALL, ns, label_selector={LABEL_TASK: name, LABEL_TCREQ: req_id}
        )
        # an empty set counts as all-completed (reference checkToolCalls
        # semantics, state_machine.go:304-317): a turn with neither content
        # nor tool calls loops straight back to the LLM
        pending = [
            tc
            for tc in tool_calls
            if tc.get("status", {}).get("status") not in ("Succeeded", "Error")
        ]
        if pending:
            # watch-driven: the owned-resource mapping requeues us on completion
            return Result(requeue_after=DEFAULT_REQUEUE)
        tool_calls.sort(key=lambda tc: tc["metadata"]["name"])
        cw = status.setdefault("contextWindow", [])
        for tc in tool_calls:
            cw.append(
                {
                    "role": "tool",
                    "content": tc.get("status", {}).get("result", ""),
                    "toolCallId": tc.get("spec", {}).get("toolCallId", ""),
                }
            )
        status.update(
            {
                "phase": TaskPhase.READY_FOR_LLM,
                "ready": True,
                "status": TaskStatusType.READY,
                "statusDetail": "All tool calls completed, ready to send tool results to LLM",
                "messageCount": len(cw),
                "error": "",
            }
        )
        self.store.record_event(task, "Normal", "AllToolCallsCompleted", "All tool calls completed")
        self.store.update_status(task)
        return Result(requeue=True)

    # ------------------------------------------------------------- terminal

    def _handle_terminal(self, task) -> Result:
        """End the trace (state_machine.go:344-361, 806-827)."""
        status = task["status"]
        if status.get("traceEnded"):
            return Result()
        sc = status.get("spanContext") or {}
        if sc.get("traceID"):
            try:
                parent = reconstruct_span_context(sc["traceID"], sc["spanID"])
                span = self.tracer.start("EndTaskSpan", parent=parent)
                ok = status.get("phase") == TaskPhase.FINAL_ANSWER
                span.set_status("OK" if ok else "ERROR", status.get("statusDetail", ""))
                span.set_attribute("task.name", task["metadata"]["name"])
                span.end()
            except ValueError:
                pass
        status["traceEnded"] = True
        self.store.update_status(task)
        return Result()

    # -------------------------------------------------------------- v1beta3

    @staticmethod
    def _v1beta3_parked(status) -> bool:
        """True when the v1beta3 thread's last completed action was a
        delivered respond_to_human round: the reference loops the task back
        to ReadyForLLM after the ToolCall succeeds (checkToolCalls has no
        special case) and the conversation effectively waits for the next
        inbound event — each new event arrives as a NEW Task carrying the
        same threadID (server.go:1384-1545), so calling the LLM again here
        would just re-answer its own delivery receipt.  Parking preserves
        the reference's phase sequence through the respond_to_human round
        while keeping the loop finite."""
        cw = status.get("contextWindow", [])
        if not cw or cw[-1].get("role") != "tool":
            return False
        last_tc_id = cw[-1].get("toolCallId", "")
        for m in reversed(cw):
            if m.get("role") != "assistant":
                continue
            for tc in m.get("toolCalls", []) or []:
                if tc.get("id") == last_tc_id:
                    return tc.get("function", {}).get("name") == "respond_to_human"
            return False
        return False

    def _v1beta3_final_answer(self, task, output: Message) -> Result:
        """Synthesize a respond_to_human ToolCall instead of FinalAnswer
        (state_machine.go:968-1067)."""
        status = task["status"]
        req_id = generate_k8s_random_string(7)
        status.update(
            {
                "output": "",
                "phase": TaskPhase.TOOL_CALLS_PENDING,
                "toolCallRequestId": req_id,
                "ready": True,
                "status": TaskStatusType.READY,
                "statusDetail": "Creating respond_to_human tool call",
                "error": "",
            }
        )
        tc_id = f"rth-{generate_k8s_random_string(7)}"
        status.setdefault("contextWindow", []).append(
            {
                "role": "assistant",
                "content": "",
                "toolCalls": [
                    {
                        "id": tc_id,
                        "function": {
                            "name": "respond_to_human",
                            "arguments": json.dumps({"content": output.content}),
                        },
                        "type": "function",
                    }
                ],
            }
        )
        status["messageCount"] = len(status["contextWindow"])
        self.store.record_event(
            task,
            "Normal",
            "V1Beta3RespondToHuman",
            "Creating respond_to_human tool call for final answer",
        )
        self.store.update_status(task)
        name = task["metadata"]["name"]
        ns = task["metadata"].get("namespace", "default")
        new_name = f"{name}-{req_id}-respond-to-human"  # state_machine.go:1022
        self.store.create(
            {
                "apiVersion": "acp.humanlayer.dev/v1alpha1",
                "kind": TOOL_CALL,
                "metadata": {
                    "name": new_name,
                    "namespace": ns,
                    "labels": {
                        LABEL_TASK: name,
                        LABEL_TCREQ: req_id,
                        V1BETA3_LABEL: "true",
                        "acp.humanlayer.dev/tool-type": "respond_to_human",
                    },
                    "ownerReferences": [owner_ref(task)],
                },
                "spec": {
                    "toolCallId": tc_id,
                    "taskRef": {"name": name},
                    "toolRef": {"name": "respond_to_human"},
                    "toolType": ToolType.HUMAN_CONTACT,
                    "arguments": json.dumps({"content": output.content}),
                },
                "status": {},
            }
        )
        self.store.record_event(
            task, "Normal", "V1Beta3ToolCallCreated", f"Created respond_to_human ToolCall {new_name}"
        )
        return Result(requeue_after=DEFAULT_REQUEUE)

    def _notify_final_result(self, task, content: str) -> None:
        try:
            ns = task["metadata"].get("namespace", "default")
            ref = task["spec"].get("contactChannelRef") or {}
            ch = self.store.get(CONTACT_CHANNEL, ref.get("name", ""), ns) or {}
            client = self.humanlayer.new_client(
                namespace=ns,
                run_id=task["metadata"]["name"],
                channel=ch.get("spec", {}),
            )
            client.notify_final_result(content)
        except Exception:
            pass
