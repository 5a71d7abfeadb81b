"""HumanLayer HTTP API client (the real wire, behind the same seam).

Parity with acp/internal/humanlayer/hlclient.go:19-222 over the generated
humanlayerapi client's endpoints:

    POST /humanlayer/v1/function_calls            (RequestApproval)
    GET  /humanlayer/v1/function_calls/{call_id}  (GetFunctionCallStatus)
    POST /humanlayer/v1/contact_requests          (RequestHumanContact)
    GET  /humanlayer/v1/contact_requests/{call_id}(GetHumanContactStatus)
    GET  /humanlayer/v1/project                   (key validation)
    GET  /humanlayer/v1/contact_channel/{id}      (channel verification)

Base URL resolution mirrors hlclient.go:23-31: explicit argument, else
``HUMANLAYER_API_BASE``, else ``https://api.humanlayer.dev``.  Requests
carry ``Authorization: Bearer <key>``; the channel config travels inside
the spec unless channel-specific auth (channelId) is used
(hlclient.go:149-206).  Wire shapes follow the generated models
(FunctionCallInput/Output, HumanContactInput/Output).

There is no egress in this deployment, so production use points
HUMANLAYER_API_BASE at a reachable endpoint and the tests run against an
in-process mock server (tests/test_humanlayer_wire.py) — the reference
does the same with httptest (contactchannel_controller_test.go:63).
"""
from __future__ import annotations

import os
import uuid
from typing import Any, Dict, Optional

from .client import FunctionCallStatus, HumanContactStatus


def resolve_api_base(explicit: str = "") -> str:
    return explicit or os.environ.get("HUMANLAYER_API_BASE", "") or "https://api.humanlayer.dev"


class HumanLayerAPIError(RuntimeError):
    def __init__(self, status_code: int, message: str):
        super().__init__(f"HumanLayer API error {status_code}: {message}")
        self.status_code = status_code


class HTTPHumanLayerClient:
    """One request/poll client (RealHumanLayerClientWrapper's role)."""

    def __init__(self, api_base: str = "", namespace: str = "default", run_id: str = "",
                 api_key: str = "", channel: Optional[Dict[str, Any]] = None,
                 channel_id: str = "", thread_id: str = "", transport=None,
                 http_client=None):
        import httpx

        self.base = resolve_api_base(api_base).rstrip("/")
        self.namespace = namespace
        self.run_id = run_id
        self.api_key = api_key
        self.channel = channel or {}
        self.channel_id = channel_id
        self.thread_id = thread_id
        self._client = http_client or httpx.Client(
            timeout=10.0, transport=transport, trust_env=False
        )

    # ------------------------------------------------------------- plumbing

    def _headers(self) -> Dict[str, str]:
        return {"Authorization": f"Bearer {self.api_key}"}

    def _check(self, r) -> Dict[str, Any]:
        if r.status_code < 200 or r.status_code >= 300:
            raise HumanLayerAPIError(r.status_code, r.text[:300])
        try:
            return r.json()
        except ValueError as e:
            raise HumanLayerAPIError(502, f"malformed response: {e}")

    def _channel_input(self) -> Optional[Dict[str, Any]]:
        """ContactChannel spec -> ContactChannelInput (hlclient.go:149-166);
        omitted under channel-specific auth."""
        if self.channel_id:
            return None
        out: Dict[str, Any] = {}
        slack = self.channel.get("slack") or {}
        if slack.get("channelOrUserID"):
            out["slack"] = {
                "channel_or_user_id": slack["channelOrUserID"],
                "context_about_channel_or_user": slack.get(
                    "contextAboutChannelOrUser", ""
                ),
            }
        email = self.channel.get("email") or {}
        if email.get("address"):
            out["email"] = {
                "address": email["address"],
                "context_about_user": email.get("contextAboutUser", ""),
            }
        return out or None

    # ------------------------------------------------------------- requests

    def request_approval(self, fn_name: str, fn_args: str, call_id: str = "") -> str:
        import json as _json

        call_id = call_id or uuid.uuid4().hex[:8]
        try:
            kwargs = _json.loads(fn_args) if isinstance(fn_args, str) else (fn_args or {})
        except ValueError:
            kwargs = {"raw": fn_args}
        spec: Dict[str, Any] = {"fn": fn_name, "kwargs": kwargs}
        ch = self._channel_input()
        if ch is not None:
            spec["channel"] = ch
        body = {"run_id": self.run_id, "call_id": call_id, "spec": spec}
        r = self._client.post(
            f"{self.base}/humanlayer/v1/function_calls",
            json=body, headers=self._headers(),
        )
        out = self._check(r)
        return out.get("call_id", call_id)

    def request_human_contact(self, message: str, call_id: str = "") -> str:
        call_id = call_id or uuid.uuid4().hex[:8]
        spec: Dict[str, Any] = {"msg": message}
        ch = self._channel_input()
        if ch is not None:
            spec["channel"] = ch
        if self.thread_id:
            spec["thread_id"] = self.thread_id
        body = {"run_id": self.run_id, "call_id": call_id, "spec": spec}
        r = self._client.post(
            f"{self.base}/humanlayer/v1/contact_requests",
            json=body, headers=self._headers(),
        )
        out = self._check(r)
        return out.get("call_id", call_id)

    # ----------------------------------------------------------------- polls

    def get_function_call_status(self, call_id: str) -> Optional[FunctionCallStatus]:
        r = self._client.get(
            f"{self.base}/humanlayer/v1/function_calls/{call_id}",
            headers=self._headers(),
        )
        if r.status_code == 404:
            return None
        out = self._check(r)
        st = out.get("status") or {}
        return FunctionCallStatus(
            requested_at=st.get("requested_at") or 0.0,
            responded_at=st.get("responded_at") or 0.0,
            approved=st.get("approved"),
            comment=st.get("comment") or "",
        )

    def get_human_contact_status(self, call_id: str) -> Optional[HumanContactStatus]:
        r = self._client.get(
            f"{self.base}/humanlayer/v1/contact_requests/{call_id}",
            headers=self._headers(),
        )
        if r.status_code == 404:
            return None
        out = self._check(r)
        st = out.get("status") or {}
        return HumanContactStatus(
            requested_at=st.get("requested_at") or 0.0,
            responded_at=st.get("responded_at") or 0.0,
            response=st.get("response"),
        )

    def notify_final_result(self, message: str) -> None:
        """Final-result delivery = a human contact that needs no reply
        (task/state_machine.go:841-860 sends through the same API)."""
        self.request_human_contact(message)


class HTTPHumanLayerClientFactory:
    """Drop-in for HumanLayerClientFactory when an API endpoint exists."""

    def __init__(self, api_base: str = "", api_key: str = "", transport=None):
        import httpx

        self.api_base = api_base
        self.api_key = api_key
        # one pooled connection shared by every per-call client
        self._http = httpx.Client(timeout=10.0, transport=transport, trust_env=False)

    def new_client(self, namespace: str = "default", run_id: str = "", api_key: str = "",
                   channel: Optional[Dict[str, Any]] = None) -> HTTPHumanLayerClient:
        return HTTPHumanLayerClient(
            api_base=self.api_base,
            namespace=namespace,
            run_id=run_id,
            api_key=api_key or self.api_key,
            channel=channel,
            http_client=self._http,
        )


# ------------------------------------------------------- channel verification


def verify_api_key_http(api_key: str, channel_id: str = "", api_base: str = "",
                        transport=None) -> Dict[str, str]:
    """contactchannel/state_machine.go:173-252: project-key validation via
    GET /humanlayer/v1/project; with channel-specific auth additionally
    verify the channel exists via GET /humanlayer/v1/contact_channel/{id}.
    Returns {projectSlug, orgSlug}; raises PermissionError / LookupError /
    HumanLayerAPIError like the reference's typed failures."""
    import httpx

    base = resolve_api_base(api_base).rstrip("/")
    client = httpx.Client(timeout=10.0, transport=transport, trust_env=False)
    try:
        r = client.get(
            f"{base}/humanlayer/v1/project",
            headers={"Authorization": f"Bearer {api_key}"},
        )
        if r.status_code == 401:
            raise PermissionError("invalid HumanLayer API key")
        if r.status_code != 200:
            raise HumanLayerAPIError(r.status_code, r.text[:200])
        body = r.json()
        if channel_id:
            rc = client.get(
                f"{base}/humanlayer/v1/contact_channel/{channel_id}",
                headers={"Authorization": f"Bearer {api_key}"},
            )
            if rc.status_code == 404:
                raise LookupError(f"channel {channel_id} not found")
            if rc.status_code == 401:
                raise PermissionError(f"invalid channel API key for channel {channel_id}")
            if rc.status_code != 200:
                raise HumanLayerAPIError(rc.status_code, rc.text[:200])
        return {
            "projectSlug": str(body.get("project_slug", "")),
            "orgSlug": str(body.get("org_slug", "")),
        }
    finally:
        client.close()
