from llmq_amd.core.config import Config, get_config
from llmq_amd.core.models import ErrorInfo, Job, QueueStats, Result, WorkerHealth

__all__ = [
    "Config",
    "get_config",
    "Job",
    "Result",
    "QueueStats",
    "WorkerHealth",
    "ErrorInfo",
]
