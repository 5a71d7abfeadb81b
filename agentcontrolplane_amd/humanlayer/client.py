"""HumanLayer client seam: approvals + human contact.

Parity with acp/internal/humanlayer/hlclient.go:55-222 (the builder-style
wrapper over the generated HumanLayer REST client) and its mock
(mock_hlclient.go:12-60).  There is no egress in this deployment, so the
production path is an in-process approval service backed by the store: an
approval/contact request becomes an ``Approval`` resource a human (or the
REST API) resolves; the poll methods read it back.  An HTTP implementation
can be slotted in behind the same interface when egress exists.

Status objects mirror FunctionCallStatus / HumanContactStatus from the
generated humanlayerapi models (approved/comment/response fields).
"""
from __future__ import annotations

import dataclasses
import threading
import time
import uuid
from typing import Any, Dict, Optional

APPROVAL = "Approval"  # store kind for pending human interactions


@dataclasses.dataclass
class FunctionCallStatus:
    requested_at: float = 0.0
    responded_at: float = 0.0
    approved: Optional[bool] = None
    comment: str = ""


@dataclasses.dataclass
class HumanContactStatus:
    requested_at: float = 0.0
    responded_at: float = 0.0
    response: Optional[str] = None


class HumanLayerClient:
    """One request/poll client instance (hlclient.go:55-69's builder collapses
    into constructor kwargs here)."""

    def __init__(self, store, namespace: str = "default", run_id: str = "", api_key: str = "",
                 channel: Optional[Dict[str, Any]] = None):
        self.store = store
        self.namespace = namespace
        self.run_id = run_id
        self.api_key = api_key
        self.channel = channel or {}

    # -------------------------------------------------------------- requests

    def request_approval(self, fn_name: str, fn_args: str, call_id: str = "") -> str:
        """hlclient.go:149-178 — returns the external call id."""
        call_id = call_id or f"fc-{uuid.uuid4().hex[:12]}"
        self.store.create(
            {
                "apiVersion": "humanlayer.dev/v1",
                "kind": APPROVAL,
                "metadata": {"name": call_id, "namespace": self.namespace},
                "spec": {
                    "type": "function_call",
                    "runId": self.run_id,
                    "fn": fn_name,
                    "kwargs": fn_args,
                    "channel": self.channel,
                    "requestedAt": time.time(),
                },
                "status": {},
            }
        )
        return call_id

    def request_human_contact(self, message: str, call_id: str = "") -> str:
        """hlclient.go:180-206."""
        call_id = call_id or f"hc-{uuid.uuid4().hex[:12]}"
        self.store.create(
            {
                "apiVersion": "humanlayer.dev/v1",
                "kind": APPROVAL,
                "metadata": {"name": call_id, "namespace": self.namespace},
                "spec": {
                    "type": "human_contact",
                    "runId": self.run_id,
                    "message": message,
                    "channel": self.channel,
                    "requestedAt": time.time(),
                },
                "status": {},
            }
        )
        return call_id

    # ----------------------------------------------------------------- polls

    def get_function_call_status(self, call_id: str) -> Optional[FunctionCallStatus]:
        obj = self.store.get(APPROVAL, call_id, self.namespace)
        if obj is None:
            return None
        st = obj.get("status", {})
        return FunctionCallStatus(
            requested_at=obj.get("spec", {}).get("requestedAt", 0.0),
            responded_at=st.get("respondedAt", 0.0),
            approved=st.get("approved"),
            comment=st.get("comment", ""),
        )

    def get_human_contact_status(self, call_id: str) -> Optional[HumanContactStatus]:
        obj = self.store.get(APPROVAL, call_id, self.namespace)
        if obj is None:
            return None
        st = obj.get("status", {})
        return HumanContactStatus(
            requested_at=obj.get("spec", {}).get("requestedAt", 0.0),
            responded_at=st.get("respondedAt", 0.0),
            response=st.get("response"),
        )

    def notify_final_result(self, message: str) -> None:
        """The async final-result notification for tasks with a
        contactChannelRef (task/state_machine.go:841-860)."""
        self.store.create(
            {
                "apiVersion": "humanlayer.dev/v1",
                "kind": APPROVAL,
                "metadata": {"name": f"notify-{uuid.uuid4().hex[:12]}", "namespace": self.namespace},
                "spec": {
                    "type": "notification",
                    "runId": self.run_id,
                    "message": message,
                    "channel": self.channel,
                    "requestedAt": time.time(),
                },
                "status": {"respondedAt": time.time()},
            }
        )


def respond_to_approval(store, call_id: str, approved: bool, comment: str = "",
                        namespace: str = "default") -> None:
    """Resolve a pending approval — what the HumanLayer SaaS does remotely."""
    obj = store.get(APPROVAL, call_id, namespace)
    if obj is None:
        raise KeyError(call_id)
    obj["status"].update(
        {"approved": approved, "comment": comment, "respondedAt": time.time()}
    )
    store.update_status(obj)


def respond_to_contact(store, call_id: str, response: str, namespace: str = "default") -> None:
    obj = store.get(APPROVAL, call_id, namespace)
    if obj is None:
        raise KeyError(call_id)
    obj["status"].update({"response": response, "respondedAt": time.time()})
    store.update_status(obj)


class HumanLayerClientFactory:
    """Factory seam injected into the ToolCall executor."""

    def __init__(self, store):
        self.store = store

    def new_client(self, namespace: str = "default", run_id: str = "", api_key: str = "",
                   channel: Optional[Dict[str, Any]] = None) -> HumanLayerClient:
        return HumanLayerClient(self.store, namespace, run_id, api_key, channel)


class MockHumanLayerClientFactory(HumanLayerClientFactory):
    """mock_hlclient.go:12-60 — approval/rejection switches for tests.

    ``auto='approve'|'reject'|None``; when set, every approval resolves
    immediately; human contacts resolve with ``auto_response`` when given.
    """

    def __init__(self, store, auto: Optional[str] = None, comment: str = "",
                 auto_response: Optional[str] = None, delay_s: float = 0.0):
        super().__init__(store)
        self.auto = auto
        self.comment = comment
        self.auto_response = auto_response
        self.delay_s = delay_s

    def new_client(self, namespace="default", run_id="", api_key="", channel=None):
        client = super().new_client(namespace, run_id, api_key, channel)
        factory = self

        class _AutoClient(HumanLayerClient):
            def request_approval(self, fn_name, fn_args, call_id=""):
                cid = HumanLayerClient.request_approval(self, fn_name, fn_args, call_id)
                if factory.auto in ("approve", "reject"):
                    def _resolve():
                        if factory.delay_s:
                            time.sleep(factory.delay_s)
                        respond_to_approval(
                            self.store, cid, factory.auto == "approve", factory.comment,
                            self.namespace,
                        )
                    threading.Thread(target=_resolve, daemon=True).start()
                return cid

            def request_human_contact(self, message, call_id=""):
                cid = HumanLayerClient.request_human_contact(self, message, call_id)
                if factory.auto_response is not None:
                    def _resolve():
                        if factory.delay_s:
                            time.sleep(factory.delay_s)
                        respond_to_contact(self.store, cid, factory.auto_response, self.namespace)
                    threading.Thread(target=_resolve, daemon=True).start()
                return cid

        auto = _AutoClient(self.store, namespace, run_id, api_key, channel)
        return auto
