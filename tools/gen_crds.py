#!/usr/bin/env python3
"""Generate the six ACP CRD manifests (config/crd/bases/).

The reference generates these with controller-gen from the Go types
(acp/api/v1alpha1/*_types.go -> acp/config/crd/bases/*.yaml); here the
schemas are generated from the same field contract as expressed by
agentcontrolplane_amd/api/types.py, so `kubectl apply` of the reference's
sample manifests (acp/config/samples/) validates against them unchanged.

Run:  python tools/gen_crds.py      (writes config/crd/bases/ + release CRD)
"""
from __future__ import annotations

import os
import sys

import yaml

GROUP = "acp.humanlayer.dev"
VERSION = "v1alpha1"

# ---------------------------------------------------------------- helpers

S = {"type": "string"}
B = {"type": "boolean"}
I = {"type": "integer"}  # noqa: E741


def obj(props, required=None, description=None, preserve=False):
    out = {"type": "object", "properties": props}
    if required:
        out["required"] = sorted(required)
    if description:
        out["description"] = description
    if preserve:
        out["x-kubernetes-preserve-unknown-fields"] = True
    return out


def arr(items):
    return {"type": "array", "items": items}


def ref():
    """LocalObjectReference — {name} (task_types.go LocalObjectReference)."""
    return obj({"name": S}, required=["name"])


def secret_key_ref():
    return obj(
        {"secretKeyRef": obj({"name": S, "key": S}, required=["name", "key"])},
        required=["secretKeyRef"],
    )


def col(name, jsonpath, typ="string", priority=None):
    c = {"jsonPath": jsonpath, "name": name, "type": typ}
    if priority is not None:
        c["priority"] = priority
    return c


STD_COLS = [
    col("Ready", ".status.ready", "boolean"),
    col("Status", ".status.status"),
    col("Detail", ".status.statusDetail", priority=1),
]

# Message / MessageToolCall (task_types.go:57-97) — the context-window
# checkpoint format; the byte format is the north-star contract.
MESSAGE_TOOL_CALL = obj(
    {
        "id": S,
        "type": S,
        "function": obj({"name": S, "arguments": S}, required=["arguments", "name"]),
    },
    required=["function", "id", "type"],
)
MESSAGE = obj(
    {
        "role": S,
        "content": S,
        "toolCalls": arr(MESSAGE_TOOL_CALL),
        "toolCallId": S,
        "name": S,
    },
    required=["role"],
)
SPAN_CONTEXT = obj({"traceID": S, "spanID": S})

# ------------------------------------------------------------------ kinds

KINDS = {
    "LLM": {
        "plural": "llms",
        "cols": [col("Provider", ".spec.provider")] + STD_COLS,
        "spec": obj(
            {
                "provider": {
                    "type": "string",
                    "enum": ["openai", "anthropic", "mistral", "google", "vertex",
                             "mock", "local"],
                },
                "apiKeyFrom": secret_key_ref(),
                "parameters": obj(
                    {
                        "model": S,
                        "baseUrl": S,
                        "temperature": S,
                        "maxTokens": I,
                        "topP": S,
                        "topK": I,
                        "frequencyPenalty": S,
                        "presencePenalty": S,
                    }
                ),
                "openai": obj({"organization": S, "apiType": S, "apiVersion": S}),
                "anthropic": obj({"anthropicBetaHeader": S}),
                "vertex": obj(
                    {"cloudProject": S, "cloudLocation": S, "maxRetries": I,
                     "timeout": I}
                ),
                "mistral": obj(
                    {"maxRetries": I, "timeout": I, "randomSeed": I, "maxTokens": I}
                ),
                "google": obj({"cloudProject": S, "cloudLocation": S}),
            },
            required=["provider"],
        ),
        "status": obj({"ready": B, "status": S, "statusDetail": S}),
    },
    "Agent": {
        "plural": "agents",
        "cols": STD_COLS,
        "spec": obj(
            {
                "llmRef": ref(),
                "system": S,
                "description": S,
                "mcpServers": arr(ref()),
                "humanContactChannels": arr(ref()),
                "subAgents": arr(ref()),
            },
            required=["llmRef", "system"],
        ),
        "status": obj(
            {
                "ready": B,
                "status": S,
                "statusDetail": S,
                "validMCPServers": arr(obj({"name": S, "tools": arr(S)})),
                "validHumanContactChannels": arr(obj({"name": S, "type": S})),
                "validSubAgents": arr(S),
            }
        ),
    },
    "Task": {
        "plural": "tasks",
        "cols": [
            col("Ready", ".status.ready", "boolean"),
            col("Status", ".status.status"),
            col("Phase", ".status.phase"),
            col("Preview", ".status.userMsgPreview"),
            col("Output", ".status.output"),
            col("Detail", ".status.statusDetail", priority=1),
            col("Error", ".status.error", priority=1),
            col("Started", ".status.startTime", "date", priority=1),
            col("Completed", ".status.completionTime", "date", priority=1),
        ],
        "spec": obj(
            {
                "agentRef": ref(),
                "userMessage": S,
                "contextWindow": arr(MESSAGE),
                "threadID": S,
                "baseURL": S,
                "channelTokenFrom": obj(
                    {"name": S, "key": S}, required=["key", "name"]
                ),
                "contactChannelRef": ref(),
            },
            required=["agentRef"],
        ),
        "status": obj(
            {
                "ready": B,
                "status": {"type": "string",
                           "enum": ["Ready", "Error", "Pending"]},
                "statusDetail": S,
                "phase": {
                    "type": "string",
                    "enum": [
                        "Initializing", "Pending", "ReadyForLLM", "SendContextWindowToLLM",
                        "ToolCallsPending", "CheckingToolCalls", "FinalAnswer",
                        "ErrorBackoff", "Failed",
                    ],
                },
                "error": S,
                "output": S,
                "contextWindow": arr(MESSAGE),
                "messageCount": I,
                "userMsgPreview": S,
                "toolCallRequestId": S,
                "spanContext": SPAN_CONTEXT,
                "startTime": S,
                "completionTime": S,
                "traceEnded": B,
            }
        ),
    },
    "ToolCall": {
        "plural": "toolcalls",
        "cols": [
            col("Phase", ".status.phase"),
            col("Task", ".spec.taskRef.name"),
            col("Tool", ".spec.toolRef.name"),
            col("Started", ".status.startTime", "date", priority=1),
            col("Completed", ".status.completionTime", "date", priority=1),
            col("Error", ".status.error", priority=1),
        ],
        "spec": obj(
            {
                "taskRef": ref(),
                "toolCallId": S,
                "toolRef": ref(),
                "toolType": S,
                "arguments": S,
            },
            required=["arguments", "taskRef", "toolCallId", "toolRef"],
        ),
        "status": obj(
            {
                "phase": S,
                "status": S,
                "statusDetail": S,
                "result": S,
                "error": S,
                "externalCallID": S,
                "startTime": S,
                "completionTime": S,
                "spanContext": SPAN_CONTEXT,
            }
        ),
    },
    "MCPServer": {
        "plural": "mcpservers",
        "cols": [
            col("Connected", ".status.connected", "boolean"),
            col("Status", ".status.status"),
            col("Detail", ".status.statusDetail", priority=1),
        ],
        "spec": obj(
            {
                "transport": {"type": "string", "enum": ["stdio", "http", "sse",
                                                         "inproc"]},
                "command": S,
                "args": arr(S),
                "url": S,
                "env": arr(
                    obj(
                        {"name": S, "value": S,
                         "valueFrom": obj({"secretKeyRef": obj(
                             {"name": S, "key": S}, required=["key", "name"])})},
                        required=["name"],
                    )
                ),
                "resources": obj(
                    {
                        "limits": obj({}, preserve=True),
                        "requests": obj({}, preserve=True),
                    }
                ),
                "approvalContactChannel": ref(),
            },
            required=["transport"],
        ),
        "status": obj(
            {
                "connected": B,
                "status": S,
                "statusDetail": S,
                "tools": arr(
                    obj(
                        {
                            "name": S,
                            "description": S,
                            "inputSchema": obj({}, preserve=True),
                        },
                        required=["name"],
                    )
                ),
            }
        ),
    },
    "ContactChannel": {
        "plural": "contactchannels",
        "cols": [col("Type", ".spec.type")] + STD_COLS,
        "spec": obj(
            {
                "type": {"type": "string", "enum": ["slack", "email"]},
                "apiKeyFrom": secret_key_ref(),
                "channelApiKeyFrom": secret_key_ref(),
                "channelId": S,
                "slack": obj(
                    {
                        "channelOrUserID": S,
                        "contextAboutChannelOrUser": S,
                        "allowedResponderIDs": arr(S),
                    },
                    required=["channelOrUserID"],
                ),
                "email": obj(
                    {"address": S, "contextAboutUser": S, "subject": S},
                    required=["address"],
                ),
            },
            required=["type"],
        ),
        "status": obj(
            {
                "ready": B,
                "status": S,
                "statusDetail": S,
                "projectSlug": S,
                "orgSlug": S,
                "verifiedChannelId": S,
            }
        ),
    },
}


def crd_for(kind: str, info: dict) -> dict:
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {
            "annotations": {"acp.humanlayer.dev/generated-by": "tools/gen_crds.py"},
            "name": f'{info["plural"]}.{GROUP}',
        },
        "spec": {
            "group": GROUP,
            "names": {
                "kind": kind,
                "listKind": f"{kind}List",
                "plural": info["plural"],
                "singular": info["plural"][:-1] if info["plural"].endswith("s") else info["plural"].lower(),
            },
            "scope": "Namespaced",
            "versions": [
                {
                    "additionalPrinterColumns": info["cols"],
                    "name": VERSION,
                    "schema": {
                        "openAPIV3Schema": obj(
                            {
                                "apiVersion": S,
                                "kind": S,
                                "metadata": {"type": "object"},
                                "spec": info["spec"],
                                "status": info["status"],
                            },
                            description=f"{kind} is the Schema for the "
                            f'{info["plural"]} API',
                        )
                    },
                    "served": True,
                    "storage": True,
                    "subresources": {"status": {}},
                }
            ],
        },
    }


def main() -> None:
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out_dir = os.path.join(root, "config", "crd", "bases")
    os.makedirs(out_dir, exist_ok=True)
    all_docs = []
    for kind, info in KINDS.items():
        doc = crd_for(kind, info)
        all_docs.append(doc)
        path = os.path.join(out_dir, f'{GROUP}_{info["plural"]}.yaml')
        with open(path, "w") as f:
            f.write("---\n")
            yaml.safe_dump(doc, f, sort_keys=True, default_flow_style=False)
        print(f"wrote {os.path.relpath(path, root)}")
    # one-shot CRD install (config/release/latest-crd.yaml in the reference)
    rel_dir = os.path.join(root, "config", "release")
    os.makedirs(rel_dir, exist_ok=True)
    with open(os.path.join(rel_dir, "latest-crd.yaml"), "w") as f:
        for doc in all_docs:
            f.write("---\n")
            yaml.safe_dump(doc, f, sort_keys=True, default_flow_style=False)
    print("wrote config/release/latest-crd.yaml")


if __name__ == "__main__":
    sys.exit(main())
