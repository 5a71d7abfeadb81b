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
        self.pod_name = pod_name
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

    #: leader-election lease constants (cmd/main.go:208-226: election ID
    #: 2994eb7c.humanlayer.dev; controller-runtime defaults 15s/10s/2s)
    LEADER_LEASE = "2994eb7c.humanlayer.dev"
    LEADER_LEASE_DURATION = 15.0
    LEADER_RETRY = 2.0

    def start(self, leader_elect: bool = False) -> "ControlPlane":
        """With ``leader_elect`` the reconcilers start only once this
        replica holds the leader Lease, and stop if leadership is lost —
        multiple pods over one store run active/passive like the
        reference's manager (cmd/main.go:208-226).  The REST server stays
        up on every replica (reads work anywhere; the reference runs its
        API leader-only, runnable.go:30-33 — here followers serve reads
        and writes land in the shared store for the leader to reconcile)."""
        if not leader_elect:
            self.manager.start()
            return self
        self.is_leader = False
        self._elect_stop = False
        import threading

        def _elect_loop() -> None:
            holder = self.pod_name
            while not self._elect_stop:
                got = self.store.acquire_lease(
                    self.LEADER_LEASE, holder, self.LEADER_LEASE_DURATION
                )
                if got and not self.is_leader:
                    self.is_leader = True
                    self.manager.start()
                elif not got and self.is_leader:
                    # leadership lost: stop reconciling immediately
                    self.is_leader = False
                    self.manager.stop()
                import time as _t

                _t.sleep(
                    self.LEADER_LEASE_DURATION / 3 if self.is_leader else self.LEADER_RETRY
                )

        self._elect_thread = threading.Thread(
            target=_elect_loop, name="acp-leader-elect", daemon=True
        )
        self._elect_thread.start()
        return self

    def stop(self) -> None:
        if getattr(self, "_elect_thread", None) is not None:
            self._elect_stop = True
            self._elect_thread.join(timeout=5)
            self._elect_thread = None
            if getattr(self, "is_leader", False):
                self.store.release_lease(self.LEADER_LEASE, self.pod_name)
        self.manager.stop()
        self.mcp.close()
        if self._owns_store and hasattr(self.store, "close"):
            self.store.close()

    def __enter__(self) -> "ControlPlane":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()
