"""Pure validation helpers.

Semantics match /root/reference/acp/internal/validation/task_validation.go:16-110.
"""
from __future__ import annotations

import secrets
from typing import Any, Dict, List, Optional

from .types import (
    CONTACT_CHANNEL,
    MESSAGE_ROLE_USER,
    VALID_MESSAGE_ROLES,
)


class ValidationError(ValueError):
    pass


class RetryableValidationError(ValidationError):
    """A dependency exists but is not ready yet — Pending + requeue instead of
    terminal Failed.  (The reference fails the task here, state_machine.go:
    463-477, which races when channel and task are created together as the
    v1beta3 handler does; waiting is strictly safer.)"""


def validate_task_message_input(
    user_message: str, context_window: Optional[List[Dict[str, Any]]]
) -> None:
    """Exactly one of userMessage / contextWindow; contextWindow needs ≥1 user
    message and only valid roles (task_validation.go:17-40)."""
    context_window = context_window or []
    if user_message and len(context_window) > 0:
        raise ValidationError("only one of userMessage or contextWindow can be provided")
    if not user_message and len(context_window) == 0:
        raise ValidationError("one of userMessage or contextWindow must be provided")
    if context_window:
        has_user = False
        for msg in context_window:
            role = msg.get("role", "")
            if role not in VALID_MESSAGE_ROLES:
                raise ValidationError(f"invalid role in contextWindow: {role}")
            if role == MESSAGE_ROLE_USER:
                has_user = True
        if not has_user:
            raise ValidationError("contextWindow must contain at least one user message")


def get_user_message_preview(
    user_message: str, context_window: Optional[List[Dict[str, Any]]]
) -> str:
    """First 50 chars of the (last) user message (task_validation.go:42-58)."""
    preview = ""
    if user_message:
        preview = user_message
    elif context_window:
        for msg in reversed(context_window):
            if msg.get("role") == MESSAGE_ROLE_USER:
                preview = msg.get("content", "")
                break
    if len(preview) > 50:
        preview = preview[:47] + "..."
    return preview


def generate_k8s_random_string(n: int) -> str:
    """k8s-style random suffix: 1-8 chars, starts with a letter
    (task_validation.go:61-87)."""
    if n < 1 or n > 8:
        n = 6
    letters = "abcdefghijklmnopqrstuvwxyz"
    alphanumeric = "abcdefghijklmnopqrstuvwxyz0123456789"
    out = [secrets.choice(letters)]
    out.extend(secrets.choice(alphanumeric) for _ in range(n - 1))
    return "".join(out)


def validate_contact_channel_ref(store, task: Dict[str, Any]) -> None:
    """Referenced ContactChannel must exist and be Ready
    (task_validation.go:90-110)."""
    ref = task.get("spec", {}).get("contactChannelRef")
    if not ref:
        return
    ns = task["metadata"].get("namespace", "default")
    cc = store.get(CONTACT_CHANNEL, ref["name"], ns)
    if cc is None:
        raise ValidationError(f'referenced ContactChannel "{ref["name"]}" not found')
    if not cc.get("status", {}).get("ready", False):
        raise RetryableValidationError(
            f'referenced ContactChannel "{ref["name"]}" is not ready '
            f'(status: {cc.get("status", {}).get("status", "")})'
        )
