from llmq_amd.workers.base import BaseWorker
from llmq_amd.workers.dummy_worker import DummyWorker


def get_engine_worker():
    """Lazy import — the engine pulls in torch."""
    from llmq_amd.workers.engine_worker import EngineWorker

    return EngineWorker


__all__ = ["BaseWorker", "DummyWorker", "get_engine_worker"]
