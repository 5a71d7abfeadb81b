"""LLM reconciler.

Parity with acp/internal/controller/llm/state_machine.go (404 LoC): validate
the provider config and API-key secret, then prove the credential with a
1-token live probe call (state_machine.go:391-401) before marking Ready.
For the ``local``/``mock`` providers the probe goes through the same
LLMClient seam (a 1-token generation against the in-process engine), so a
Ready local LLM means the engine actually answered.
"""
from __future__ import annotations

from typing import Optional

from ..api.types import LLM, SECRET, Message
from ..llmclient.base import LLMRequestError
from ..llmclient.factory import KNOWN_PROVIDERS, LLMClientFactory
from .manager import Reconciler, Result


class LLMReconciler(Reconciler):
    kind = LLM
    workers = 2

    def __init__(self, store, llm_client_factory: Optional[LLMClientFactory] = None,
                 probe: bool = True):
        super().__init__(store)
        self.factory = llm_client_factory or LLMClientFactory()
        self.probe = probe

    def reconcile(self, name: str, namespace: str) -> Result:
        llm = self.store.get(LLM, name, namespace)
        if llm is None:
            return Result()
        spec = llm.get("spec", {})
        status = llm.setdefault("status", {})

        def fail(detail: str, reason: str = "ValidationFailed", requeue: float = 30.0) -> Result:
            status.update({"ready": False, "status": "Error", "statusDetail": detail})
            self.store.record_event(llm, "Warning", reason, detail)
            self.store.update_status(llm)
            return Result(requeue_after=requeue)

        provider = spec.get("provider", "")
        if provider not in KNOWN_PROVIDERS:
            return fail(f"unsupported provider: {provider!r}", requeue=0)

        # API-key secret resolution (state_machine.go:185-240): required for
        # remote providers; local/mock need none.
        api_key = ""
        if provider not in ("mock", "local"):
            src = spec.get("apiKeyFrom", {}) or {}
            ref = src.get("secretKeyRef", {}) or {}
            if not ref.get("name"):
                return fail("apiKeyFrom.secretKeyRef is required for remote providers")
            secret = self.store.get(SECRET, ref["name"], namespace)
            if secret is None:
                return fail(f'secret "{ref["name"]}" not found', "SecretFetchFailed")
            data = secret.get("spec", {}).get("data", {}) or secret.get("data", {})
            api_key = str(data.get(ref.get("key", ""), ""))
            if not api_key:
                return fail(f'key "{ref.get("key")}" empty in secret "{ref["name"]}"')

        # provider-specific config shape checks (state_machine.go:244-404)
        if provider == "vertex":
            v = spec.get("vertex") or {}
            if not v.get("cloudProject") or not v.get("cloudLocation"):
                return fail("vertex provider requires cloudProject and cloudLocation")

        if self.probe:
            try:
                client = self.factory.create_client(llm, api_key)
                # the 1-token probe (GenerateFromSinglePrompt with MaxTokens(1))
                client.send_request(
                    [Message(role="user", content="ping")], tools=[]
                )
            except LLMRequestError as e:
                retry = 0.0 if 400 <= e.status_code < 500 else 30.0
                return fail(f"probe failed: {e}", "ValidationFailed", requeue=retry)
            except Exception as e:  # engine not up yet, etc.
                return fail(f"probe failed: {e}", "ValidationFailed", requeue=5.0)

        status.update({"ready": True, "status": "Ready", "statusDetail": "LLM validated"})
        self.store.record_event(llm, "Normal", "ValidationSucceeded", "LLM validated successfully")
        self.store.update_status(llm)
        return Result()
