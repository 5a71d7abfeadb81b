"""ToolCall reconciler + executor.

Parity with acp/internal/controller/toolcall/ (state_machine.go 403 LoC +
executor.go 401 LoC): flat phase dispatch (state_machine.go:53-70) —

  "" → span init → Pending/Pending → Pending/Ready → checkApproval
     → execute | AwaitingHumanApproval (poll) → ReadyToExecuteApprovedTool
     → Succeeded | Failed | ToolCallRejected

The executor routes on spec.toolType (executor.go:36-54):
- MCP:             mcpManager.call_tool via the ``server__tool`` convention
- DelegateToAgent: idempotently create child Task ``delegate-<tc>-<agent>``
  labeled ``acp.humanlayer.dev/parent-toolcall`` (executor.go:176-242), then
  wait for it (watch-driven here: this controller owns Tasks by that label)
- HumanContact:    HumanLayer request + poll (executor.go:332-401)

A rejected approval is phase ToolCallRejected but **status Succeeded** with
result "Rejected: <comment>" so the rejection feeds back into the LLM loop
(state_machine.go:153-160).
"""
from __future__ import annotations

import json
import time
from typing import Optional

from ..api.types import (
    CONTACT_CHANNEL,
    MCP_SERVER,
    TASK,
    TOOL_CALL,
    TaskPhase,
    ToolCallPhase,
    ToolCallStatusType,
    ToolType,
)
from ..tracing import get_tracer, reconstruct_span_context
from .manager import Reconciler, Result

APPROVAL_POLL = 5.0        # toolcall/state_machine.go:118-146
APPROVAL_ERR_POLL = 15.0
SUBAGENT_POLL = 5.0
PARENT_TC_LABEL = "acp.humanlayer.dev/parent-toolcall"


class ToolCallReconciler(Reconciler):
    kind = TOOL_CALL
    owns = (TASK,)
    workers = 8

    def __init__(self, store, mcp_manager=None, humanlayer_factory=None):
        super().__init__(store)
        self.mcp = mcp_manager
        self.humanlayer = humanlayer_factory
        self.tracer = get_tracer()

    def map_owned(self, ev):
        # child Task completion requeues the delegating ToolCall
        labels = ev.obj.get("metadata", {}).get("labels", {}) or {}
        parent = labels.get(PARENT_TC_LABEL)
        if parent:
            return parent, ev.obj["metadata"].get("namespace", "default")
        return None

    # ------------------------------------------------------------- dispatch

    def reconcile(self, name: str, namespace: str) -> Result:
        tc = self.store.get(TOOL_CALL, name, namespace)
        if tc is None:
            return Result()
        status = tc.setdefault("status", {})
        phase = status.get("phase", "")
        if phase == "":
            return self._initialize(tc)
        if phase == ToolCallPhase.PENDING and status.get("status") == ToolCallStatusType.PENDING:
            return self._setup(tc)
        if phase == ToolCallPhase.PENDING:
            return self._check_approval(tc)
        if phase == ToolCallPhase.AWAITING_HUMAN_APPROVAL:
            return self._wait_for_approval(tc)
        if phase == ToolCallPhase.READY_TO_EXECUTE_APPROVED_TOOL:
            return self._execute(tc)
        if phase == ToolCallPhase.AWAITING_SUB_AGENT:
            return self._wait_for_sub_agent(tc)
        if phase == ToolCallPhase.AWAITING_HUMAN_INPUT:
            return self._wait_for_human_input(tc)
        if phase in (
            ToolCallPhase.ERROR_REQUESTING_HUMAN_APPROVAL,
            ToolCallPhase.ERROR_REQUESTING_HUMAN_INPUT,
        ):
            return self._check_approval(tc)  # retry the request
        return Result()  # terminal

    # ---------------------------------------------------------------- steps

    def _initialize(self, tc) -> Result:
        """Span init + Pending/Pending (state_machine.go:75-82, 308-319)."""
        status = tc["status"]
        ns = tc["metadata"].get("namespace", "default")
        parent_task = self.store.get(TASK, tc.get("spec", {}).get("taskRef", {}).get("name", ""), ns)
        parent_ctx = None
        if parent_task:
            sc = parent_task.get("status", {}).get("spanContext") or {}
            if sc.get("traceID"):
                try:
                    parent_ctx = reconstruct_span_context(sc["traceID"], sc["spanID"])
                except ValueError:
                    parent_ctx = None
        span = self.tracer.start(
            "ToolCall", parent=parent_ctx, attributes={"toolcall.name": tc["metadata"]["name"]}
        )
        status.update(
            {
                "phase": ToolCallPhase.PENDING,
                "status": ToolCallStatusType.PENDING,
                "statusDetail": "Initializing",
                "spanContext": {"traceID": span.trace_id, "spanID": span.span_id},
                "startTime": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            }
        )
        span.end()
        self.store.update_status(tc)
        return Result(requeue=True)

    def _setup(self, tc) -> Result:
        """Pending/Pending → Pending/Ready (state_machine.go:84-89)."""
        tc["status"].update(
            {"status": ToolCallStatusType.READY, "statusDetail": "Ready for execution"}
        )
        self.store.update_status(tc)
        return Result(requeue=True)

    def _approval_channel(self, tc) -> Optional[dict]:
        """MCP tool → owning MCPServer → spec.approvalContactChannel
        (executor.go:57-82)."""
        if tc.get("spec", {}).get("toolType") != ToolType.MCP:
            return None
        tool_name = tc["spec"].get("toolRef", {}).get("name", "")
        if "__" not in tool_name:
            return None
        server_name = tool_name.split("__", 1)[0]
        ns = tc["metadata"].get("namespace", "default")
        srv = self.store.get(MCP_SERVER, server_name, ns)
        if srv is None:
            return None
        acc = srv.get("spec", {}).get("approvalContactChannel")
        if not acc:
            return None
        return self.store.get(CONTACT_CHANNEL, acc.get("name", ""), ns)

    def _check_approval(self, tc) -> Result:
        """state_machine.go:91-119."""
        channel = self._approval_channel(tc)
        if channel is None:
            return self._execute(tc)
        # approval required → request it
        status = tc["status"]
        ns = tc["metadata"].get("namespace", "default")
        try:
            client = self.humanlayer.new_client(
                namespace=ns, run_id=tc["metadata"]["name"], channel=channel.get("spec", {})
            )
            call_id = client.request_approval(
                tc["spec"].get("toolRef", {}).get("name", ""),
                tc["spec"].get("arguments", ""),
            )
        except Exception as e:
            status.update(
                {
                    "phase": ToolCallPhase.ERROR_REQUESTING_HUMAN_APPROVAL,
                    "status": ToolCallStatusType.ERROR,
                    "statusDetail": f"approval request failed: {e}",
                    "error": str(e),
                }
            )
            self.store.record_event(tc, "Warning", "ApprovalRequestFailed", str(e))
            self.store.update_status(tc)
            return Result(requeue_after=APPROVAL_ERR_POLL)
        status.update(
            {
                "phase": ToolCallPhase.AWAITING_HUMAN_APPROVAL,
                "status": ToolCallStatusType.PENDING,
                "statusDetail": "Waiting for human approval",
                "externalCallID": call_id,
            }
        )
        self.store.record_event(tc, "Normal", "AwaitingHumanApproval", "Approval requested")
        self.store.update_status(tc)
        return Result(requeue_after=APPROVAL_POLL)

    def _wait_for_approval(self, tc) -> Result:
        """Poll approval (state_machine.go:121-161)."""
        status = tc["status"]
        ns = tc["metadata"].get("namespace", "default")
        client = self.humanlayer.new_client(namespace=ns)
        fc = client.get_function_call_status(status.get("externalCallID", ""))
        if fc is None or fc.approved is None:
            return Result(requeue_after=APPROVAL_POLL)
        if fc.approved:
            status.update(
                {
                    "phase": ToolCallPhase.READY_TO_EXECUTE_APPROVED_TOOL,
                    "status": ToolCallStatusType.READY,
                    "statusDetail": "Approved, ready to execute",
                }
            )
            self.store.record_event(tc, "Normal", "Approved", "Tool call approved")
            self.store.update_status(tc)
            return Result(requeue=True)
        # rejection feeds back to the LLM as a Succeeded result (153-160)
        status.update(
            {
                "phase": ToolCallPhase.TOOL_CALL_REJECTED,
                "status": ToolCallStatusType.SUCCEEDED,
                "statusDetail": "Tool call rejected",
                "result": f"Rejected: {fc.comment}" if fc.comment else "Rejected",
                "completionTime": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            }
        )
        self.store.record_event(tc, "Normal", "Rejected", "Tool call rejected by human")
        self.store.update_status(tc)
        return Result()

    # -------------------------------------------------------------- execute

    def _execute(self, tc) -> Result:
        """Dispatch on toolType (state_machine.go:163-216, executor.go:36-54)."""
        tool_type = tc.get("spec", {}).get("toolType", "")
        if tool_type == ToolType.DELEGATE_TO_AGENT:
            return self._execute_delegate(tc)
        if tool_type == ToolType.HUMAN_CONTACT:
            return self._execute_human_contact(tc)
        return self._execute_external(tc) or self._execute_mcp(tc)

    def _execute_external(self, tc) -> Optional[Result]:
        """Optional external-API path (externalAPI/main.go:32-67 is unused
        scaffolding in the reference; here a ToolCall whose toolRef matches a
        registered client executes through the registry instead of MCP).
        Returns None when the tool is not registered."""
        from ..external_api import DEFAULT_REGISTRY

        tool_name = tc["spec"].get("toolRef", {}).get("name", "")
        if not DEFAULT_REGISTRY.has(tool_name):
            return None
        try:
            args = json.loads(tc["spec"].get("arguments", "") or "{}")
        except json.JSONDecodeError as e:
            return self._fail(tc, f"invalid arguments JSON: {e}")
        try:
            client = DEFAULT_REGISTRY.get_client(tool_name)
            result = client.call(args)
        except Exception as e:
            return self._fail(tc, str(e))
        return self._finish(tc, result)

    def _finish(self, tc, result: str) -> Result:
        status = tc["status"]
        status.update(
            {
                "phase": ToolCallPhase.SUCCEEDED,
                "status": ToolCallStatusType.SUCCEEDED,
                "statusDetail": "Tool executed successfully",
                "result": result,
                "error": "",
                "completionTime": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            }
        )
        self.store.record_event(tc, "Normal", "ExecutionSucceeded", "Tool executed successfully")
        self.store.update_status(tc)
        return Result()

    def _fail(self, tc, err: str) -> Result:
        status = tc["status"]
        status.update(
            {
                "phase": ToolCallPhase.FAILED,
                "status": ToolCallStatusType.ERROR,
                "statusDetail": f"Tool execution failed: {err}",
                "error": err,
                "completionTime": time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime()),
            }
        )
        self.store.record_event(tc, "Warning", "ExecutionFailed", err)
        self.store.update_status(tc)
        return Result()

    def _execute_mcp(self, tc) -> Result:
        """executor.go:164-174."""
        if self.mcp is None:
            return self._fail(tc, "no MCP manager configured")
        tool_name = tc["spec"].get("toolRef", {}).get("name", "")
        server, tool = self.mcp.find_server_for_tool(tool_name)
        if server is None:
            return self._fail(tc, f"no MCP server found for tool {tool_name!r}")
        try:
            args = json.loads(tc["spec"].get("arguments", "") or "{}")
        except json.JSONDecodeError as e:
            return self._fail(tc, f"invalid arguments JSON: {e}")
        try:
            result = self.mcp.call_tool(server, tool, args)
        except Exception as e:
            return self._fail(tc, str(e))
        return self._finish(tc, result)

    def _execute_delegate(self, tc) -> Result:
        """executor.go:176-242: idempotent child Task creation."""
        tool_name = tc["spec"].get("toolRef", {}).get("name", "")
        agent_name = tool_name.split("__", 1)[1] if "__" in tool_name else tool_name
        try:
            args = json.loads(tc["spec"].get("arguments", "") or "{}")
        except json.JSONDecodeError as e:
            return self._fail(tc, f"invalid arguments JSON: {e}")
        message = args.get("message")
        if not isinstance(message, str) or not message:
            return self._fail(tc, "missing or invalid 'message' argument")
        ns = tc["metadata"].get("namespace", "default")
        child_name = f'delegate-{tc["metadata"]["name"]}-{agent_name}'
        if len(child_name) > 63:
            child_name = child_name[:55] + "-" + child_name[-7:]
        existing = self.store.get(TASK, child_name, ns)
        if existing is None:
            try:
                self.store.create(
                    {
                        "apiVersion": "acp.humanlayer.dev/v1alpha1",
                        "kind": TASK,
                        "metadata": {
                            "name": child_name,
                            "namespace": ns,
                            "labels": {PARENT_TC_LABEL: tc["metadata"]["name"]},
                        },
                        "spec": {"agentRef": {"name": agent_name}, "userMessage": message},
                        "status": {},
                    }
                )
            except Exception:
                pass  # concurrent create — idempotent
        elif (existing.get("metadata", {}).get("labels", {}) or {}).get(PARENT_TC_LABEL) != tc[
            "metadata"
        ]["name"]:
            return self._fail(tc, f"task {child_name} exists but is not a child of this toolcall")
        tc["status"].update(
            {
                "phase": ToolCallPhase.AWAITING_SUB_AGENT,
                "status": ToolCallStatusType.PENDING,
                "statusDetail": f"Delegated to agent {agent_name} via task {child_name}",
            }
        )
        self.store.record_event(
            tc, "Normal", "SubAgentDelegated", f"Created child task {child_name}"
        )
        self.store.update_status(tc)
        return Result(requeue_after=SUBAGENT_POLL)

    def _wait_for_sub_agent(self, tc) -> Result:
        """state_machine.go:218-267 — watch-driven here via owns=(Task,)."""
        ns = tc["metadata"].get("namespace", "default")
        children = self.store.list(TASK, ns, label_selector={PARENT_TC_LABEL: tc["metadata"]["name"]})
        if not children:
            return Result(requeue_after=SUBAGENT_POLL)
        child = children[0]
        phase = child.get("status", {}).get("phase", "")
        if phase == TaskPhase.FINAL_ANSWER:
            return self._finish(tc, child.get("status", {}).get("output", ""))
        if phase == TaskPhase.FAILED:
            return self._fail(
                tc, f'sub-agent task failed: {child.get("status", {}).get("error", "")}'
            )
        return Result(requeue_after=SUBAGENT_POLL)

    def _execute_human_contact(self, tc) -> Result:
        """executor.go:269-401 (incl. the v1beta3 respond_to_human path)."""
        if self.humanlayer is None:
            return self._fail(tc, "no HumanLayer client configured")
        tool_name = tc["spec"].get("toolRef", {}).get("name", "")
        try:
            args = json.loads(tc["spec"].get("arguments", "") or "{}")
        except json.JSONDecodeError as e:
            return self._fail(tc, f"invalid arguments JSON: {e}")
        ns = tc["metadata"].get("namespace", "default")
        channel_spec = {}
        if "__" in tool_name:
            ch = self.store.get(CONTACT_CHANNEL, tool_name.split("__", 1)[0], ns)
            channel_spec = (ch or {}).get("spec", {})
        message = args.get("message") or args.get("content") or ""
        if tool_name == "respond_to_human":
            # v1beta3: the final answer itself is the human contact
            client = self.humanlayer.new_client(
                namespace=ns, run_id=tc["metadata"]["name"], channel=channel_spec
            )
            try:
                client.notify_final_result(message)
            except Exception as e:
                return self._fail(tc, str(e))
            # executor.go:400 result format
            call_id = tc["spec"].get("toolCallId", "")
            return self._finish(tc, f"Response sent to human, call ID: {call_id}")
        try:
            client = self.humanlayer.new_client(
                namespace=ns, run_id=tc["metadata"]["name"], channel=channel_spec
            )
            call_id = client.request_human_contact(message)
        except Exception as e:
            tc["status"].update(
                {
                    "phase": ToolCallPhase.ERROR_REQUESTING_HUMAN_INPUT,
                    "status": ToolCallStatusType.ERROR,
                    "statusDetail": f"human contact request failed: {e}",
                    "error": str(e),
                }
            )
            self.store.record_event(tc, "Warning", "HumanContactRequestFailed", str(e))
            self.store.update_status(tc)
            return Result(requeue_after=APPROVAL_ERR_POLL)
        tc["status"].update(
            {
                "phase": ToolCallPhase.AWAITING_HUMAN_INPUT,
                "status": ToolCallStatusType.PENDING,
                "statusDetail": "Waiting for human response",
                "externalCallID": call_id,
            }
        )
        self.store.record_event(tc, "Normal", "AwaitingHumanInput", "Human contact requested")
        self.store.update_status(tc)
        return Result(requeue_after=APPROVAL_POLL)

    def _wait_for_human_input(self, tc) -> Result:
        """state_machine.go:269-306."""
        ns = tc["metadata"].get("namespace", "default")
        client = self.humanlayer.new_client(namespace=ns)
        hc = client.get_human_contact_status(tc["status"].get("externalCallID", ""))
        if hc is None or hc.response is None:
            return Result(requeue_after=APPROVAL_POLL)
        return self._finish(tc, hc.response)
