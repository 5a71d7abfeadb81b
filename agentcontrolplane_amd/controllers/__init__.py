from .manager import ControllerManager, Reconciler, Result  # noqa: F401
