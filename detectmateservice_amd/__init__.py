"""detectmate-mi355x: MI355X-native streaming log-anomaly-detection framework.

Public surface (re-exports mirror the reference's
/root/reference/src/service/__init__.py:1-12).
"""
from .metadata import __version__
from .settings import ServiceSettings, TlsInputConfig, TlsOutputConfig
from .core import Service
from .engine.engine import Engine, EngineException, Processor
from .engine.sockets import (
    EngineSocket,
    PairDialer,
    PairListener,
    PairSocketFactory,
    RecvTimeout,
)
from .components.base import (
    BufferMode,
    CoreComponent,
    CoreConfig,
    CoreDetector,
    CoreDetectorConfig,
)

__all__ = [
    "__version__",
    "Service",
    "ServiceSettings",
    "TlsInputConfig",
    "TlsOutputConfig",
    "Engine",
    "EngineException",
    "Processor",
    "EngineSocket",
    "PairListener",
    "PairDialer",
    "PairSocketFactory",
    "RecvTimeout",
    "CoreComponent",
    "CoreConfig",
    "CoreDetector",
    "CoreDetectorConfig",
    "BufferMode",
]
