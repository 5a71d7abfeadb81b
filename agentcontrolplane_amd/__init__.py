"""agentcontrolplane_amd — an MI355X-native agent control plane.

A from-scratch rebuild of the capabilities of humanlayer/agentcontrolplane
(reference: /root/reference, a Go/kubebuilder operator) as a native stack for
AMD MI355X (gfx950) nodes:

- ``api``         resource schemas matching the acp.humanlayer.dev/v1alpha1 CRDs
                  (reference: acp/api/v1alpha1/*_types.go)
- ``store``       durable resource store with watches, leases and events —
                  the role etcd + the apiserver play for the reference
- ``controllers`` state-machine reconcilers for LLM / Agent / Task / ToolCall /
                  MCPServer / ContactChannel (reference: acp/internal/controller/*)
- ``llmclient``   the LLMClient seam (reference: acp/internal/llmclient) with a
                  mock provider and the *local* provider backed by the in-process
                  MI355X inference engine
- ``engine``      paged-KV continuous-batching inference engine (new; no
                  reference counterpart — replaces remote LLM HTTP calls)
- ``models``      Llama / Mixtral model definitions running on hand-written
                  CDNA4 HIP kernels
- ``ops``         HIP kernel bindings + fp32 torch reference implementations
- ``parallel``    tensor/expert parallelism over RCCL/xGMI
- ``mcp``         MCP server manager (stdio subprocess + http)
- ``humanlayer``  human approval / contact client seam
- ``server``      north-bound REST API (/v1/tasks, /v1/agents, /v1/beta3/events)
"""

__version__ = "0.1.0"

GROUP = "acp.humanlayer.dev"
VERSION = "v1alpha1"
API_VERSION = f"{GROUP}/{VERSION}"
