"""Process wiring — the cmd/main.go equivalent.

The reference's main() (acp/cmd/main.go:68-327) builds one manager with six
reconcilers sharing a single MCPManager, an LLM client factory, otel, and
the REST server.  ``ControlPlane`` assembles the same graph in-process:

    cp = ControlPlane(engine=my_engine)       # engine optional
    cp.start()
    ... cp.store / cp.manager / cp.rest_app ...
    cp.stop()
"""
from __future__ import annotations

from typing import Optional

from .controllers.agent import AgentReconciler
from .controllers.contactchannel import ContactChannelReconciler
from .controllers.llm import LLMReconciler
from .controllers.manager import ControllerManager
from .controllers.mcpserver import MCPServerReconciler
from .controllers.task import TaskReconciler
from .controllers.toolcall import ToolCallReconciler
from .humanlayer import HumanLayerClientFactory, MockHumanLayerClientFactory
from .llmclient.factory import LLMClientFactory
from .mcp.manager import MCPServerManager
from .store import ResourceStore
from .tracing import get_tracer


class ControlPlane:
    def __init__(
        self,
        wal_path: Optional[str] = None,
        engine=None,
        llm_client_factory: Optional[LLMClientFactory] = None,
        humanlayer_factory=None,
        auto_approve: Optional[str] = None,
        llm_probe: bool = True,
        pod_name: str = "acp-controller-0",
        fsync: str = "interval",
        store: Optional[ResourceStore] = None,
    ):
        # an injected store lets several ControlPlane replicas share one
        # backing store (the reference's multi-pod deployment over one etcd)
        self.store = store if store is not None else ResourceStore(wal_path=wal_path, fsync=fsync)
        self._owns_store = store is None
        self.tracer = get_tracer()
        self.engine = engine
        self.mcp = MCPServerManager(self.store)
        if humanlayer_factory is not None:
            self.humanlayer = humanlayer_factory
        elif auto_approve is not None:
            self.humanlayer = MockHumanLayerClientFactory(self.store, auto=auto_approve)
        else:
            self.humanlayer = HumanLayerClientFactory(self.store)

        if llm_client_factory is not None:
            self.llm_factory = llm_client_factory
        else:
            engine_provider = None
            if engine is not None:
                from .llmclient.local import LocalEngineClient

                def engine_provider(llm):
                    return LocalEngineClient(engine, llm)

            self.llm_factory = LLMClientFactory(engine_provider=engine_provider)

        self.manager = ControllerManager(self.store)
        self.manager.register(LLMReconciler(self.store, self.llm_factory, probe=llm_probe))
        self.manager.register(AgentReconciler(self.store))
        self.manager.register(ContactChannelReconciler(self.store))
        self.manager.register(MCPServerReconciler(self.store, self.mcp))
        self.task_reconciler = TaskReconciler(
            self.store, self.llm_factory, self.mcp, self.humanlayer, pod_name=pod_name
        )
        self.manager.register(self.task_reconciler)
        self.manager.register(ToolCallReconciler(self.store, self.mcp, self.humanlayer))
        self._rest_app = None

    # ------------------------------------------------------------------ REST

    @property
    def rest_app(self):
        if self._rest_app is None:
            from .server.rest import build_app

            self._rest_app = build_app(self.store, self.manager, engine=self.engine)
        return self._rest_app

    # ------------------------------------------------------------- lifecycle

    def start(self) -> "ControlPlane":
        self.manager.start()
        return self

    def stop(self) -> None:
        self.manager.stop()
        self.mcp.close()
        if self._owns_store:
            self.store.close()

    def __enter__(self) -> "ControlPlane":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
