"""ContactChannel reconciler.

Parity with acp/internal/controller/contactchannel/state_machine.go (402 LoC):
validate channel type + config shape, mutual-exclusion rules between
``apiKeyFrom`` and ``channelApiKeyFrom``+``channelId``
(state_machine.go:301-327), resolve the API key secret, and verify the key
against the HumanLayer API (state_machine.go:214-252).  The verification
endpoint is an injectable callable (the reference makes it a package var so
tests can point it at httptest; the default here accepts keys prefixed
``sk-`` or ``hl-``, standing in for the live API since there is no egress).
"""
from __future__ import annotations

from typing import Callable, Dict, Optional

from ..api.types import CONTACT_CHANNEL, SECRET
from .manager import Reconciler, Result


def default_verify_api_key(api_key: str, channel_id: str = "") -> Dict[str, str]:
    """Key verification against the HumanLayer API when an endpoint is
    configured, offline stand-in otherwise.

    The reference GETs ``humanLayerAPIURL`` (a package var defaulting to
    the live SaaS, contactchannel_controller.go:36) and, for
    channel-specific auth, the channel endpoint
    (state_machine.go:173-252).  Here: with ``HUMANLAYER_API_BASE`` set
    the same HTTP verification runs (humanlayer/wire.py,
    test-overridable exactly like the reference's httptest pattern);
    without it — this deployment has no egress — a prefix check stands
    in so channels can exist offline."""
    import os

    if os.environ.get("HUMANLAYER_API_BASE"):
        from ..humanlayer.wire import verify_api_key_http

        return verify_api_key_http(api_key, channel_id)
    if api_key.startswith(("sk-", "hl-")):
        return {"projectSlug": "local-project", "orgSlug": "local-org"}
    raise PermissionError("invalid HumanLayer API key")


class ContactChannelReconciler(Reconciler):
    kind = CONTACT_CHANNEL
    workers = 1

    def __init__(self, store, verify_api_key: Optional[Callable] = None):
        super().__init__(store)
        self.verify_api_key = verify_api_key or default_verify_api_key

    def reconcile(self, name: str, namespace: str) -> Result:
        ch = self.store.get(CONTACT_CHANNEL, name, namespace)
        if ch is None:
            return Result()
        spec = ch.get("spec", {})
        status = ch.setdefault("status", {})

        def fail(detail: str, requeue: float = 30.0) -> Result:
            status.update({"ready": False, "status": "Error", "statusDetail": detail})
            self.store.record_event(ch, "Warning", "ValidationFailed", detail)
            self.store.update_status(ch)
            return Result(requeue_after=requeue)

        ctype = spec.get("type", "")
        if ctype not in ("slack", "email"):
            return fail(f"unsupported channel type: {ctype!r}", requeue=0)
        if ctype == "slack" and not (spec.get("slack") or {}).get("channelOrUserID", ""):
            return fail("slack config requires channelOrUserID", requeue=0)
        if ctype == "email" and not (spec.get("email") or {}).get("address", ""):
            return fail("email config requires address", requeue=0)

        # field mutual exclusion (state_machine.go:301-327)
        has_api_key = bool(spec.get("apiKeyFrom"))
        has_channel_key = bool(spec.get("channelApiKeyFrom"))
        channel_id = spec.get("channelId", "")
        if has_api_key and has_channel_key:
            return fail("apiKeyFrom and channelApiKeyFrom are mutually exclusive", requeue=0)
        if has_channel_key and not channel_id:
            return fail("channelApiKeyFrom requires channelId", requeue=0)
        if not has_api_key and not has_channel_key:
            return fail("one of apiKeyFrom or channelApiKeyFrom is required", requeue=0)

        src = spec.get("apiKeyFrom") or spec.get("channelApiKeyFrom") or {}
        ref = src.get("secretKeyRef", {}) or {}
        secret = self.store.get(SECRET, ref.get("name", ""), namespace)
        if secret is None:
            return fail(f'secret "{ref.get("name")}" not found')
        data = secret.get("spec", {}).get("data", {}) or secret.get("data", {})
        api_key = str(data.get(ref.get("key", ""), ""))
        if not api_key:
            return fail(f'key "{ref.get("key")}" empty in secret "{ref.get("name")}"')

        try:
            slugs = self.verify_api_key(api_key, channel_id)
        except Exception as e:
            return fail(f"API key verification failed: {e}")

        status.update(
            {
                "ready": True,
                "status": "Ready",
                "statusDetail": "ContactChannel validated",
                "projectSlug": slugs.get("projectSlug", ""),
                "orgSlug": slugs.get("orgSlug", ""),
            }
        )
        if channel_id:
            status["verifiedChannelId"] = channel_id
        self.store.record_event(ch, "Normal", "ValidationSucceeded", "ContactChannel validated")
        self.store.update_status(ch)
        return Result()
