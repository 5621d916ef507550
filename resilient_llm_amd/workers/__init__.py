from .base import (  # noqa: F401
    GenerationRequest, GenerationChunk, GenerationResult, Worker,
    WorkerError, WorkerThrottled, WorkerDead, WorkerRegistry,
)
from .stub import StubWorker  # noqa: F401
from .engine_worker import EngineWorker  # noqa: F401
