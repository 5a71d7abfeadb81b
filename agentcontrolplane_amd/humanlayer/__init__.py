from .client import (  # noqa: F401
    FunctionCallStatus,
    HumanContactStatus,
    HumanLayerClient,
    HumanLayerClientFactory,
    MockHumanLayerClientFactory,
)
