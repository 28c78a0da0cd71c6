"""Worker process entry points (reference parity: llmq/cli/worker.py:9-250).

Each launcher builds a worker and runs its asyncio loop. The engine worker
replaces the reference's vLLM worker (run_vllm_worker, cli/worker.py:9-57):
GPU selection via HIP_VISIBLE_DEVICES, tensor parallelism over RCCL/xGMI.
"""

from __future__ import annotations

import asyncio
import logging
import sys
from typing import Optional

from llmq_amd.core.pipeline import PipelineConfig
from llmq_amd.utils.logging import setup_logging

logger = logging.getLogger(__name__)


def run_engine_worker(
    model: str,
    queue_name: str,
    tensor_parallel_size: Optional[int] = None,
    max_num_seqs: Optional[int] = None,
    max_model_len: Optional[int] = None,
    prefetch: Optional[int] = None,
    pipeline: Optional[PipelineConfig] = None,
    stage_name: Optional[str] = None,
    stage_config: Optional[dict] = None,
    engine_overrides: Optional[dict] = None,
) -> None:
    setup_logging(json_output=True)
    from llmq_amd.workers.engine_worker import EngineWorker

    worker = EngineWorker(
        queue_name,
        model=model,
        tensor_parallel_size=tensor_parallel_size,
        max_num_seqs=max_num_seqs,
        max_model_len=max_model_len,
        prefetch=prefetch,
        pipeline=pipeline,
        stage_name=stage_name,
        stage_config=stage_config or {},
        engine_overrides=engine_overrides,
    )
    asyncio.run(worker.run())


def _dp_child(rank: int, gpus_per_replica: int, kwargs: dict) -> None:
    import os

    first = rank * gpus_per_replica
    visible = ",".join(str(first + i) for i in range(gpus_per_replica))
    os.environ["HIP_VISIBLE_DEVICES"] = visible
    os.environ.setdefault("CUDA_VISIBLE_DEVICES", visible)
    run_engine_worker(**kwargs)


def run_engine_worker_dp(
    model: str,
    queue_name: str,
    data_parallel_size: int,
    tensor_parallel_size: Optional[int] = None,
    **kwargs,
) -> None:
    """N independent engine replicas on one node, each pinned to its own GPU
    slice, all competing on QUEUE_NAME (queue-level data parallelism — the
    reference's --data-parallel-size, main.py:433-439, realised the way its
    production scripts actually scale: one process per replica,
    run_dutch_nemotron.slurm:50-74)."""
    import multiprocessing as mp

    setup_logging()
    tp = tensor_parallel_size or 1
    ctx = mp.get_context("spawn")
    child_kwargs = dict(
        model=model, queue_name=queue_name,
        tensor_parallel_size=tensor_parallel_size, **kwargs,
    )
    procs = [
        ctx.Process(target=_dp_child, args=(r, tp, child_kwargs), daemon=False)
        for r in range(data_parallel_size)
    ]
    for proc in procs:
        proc.start()
    logger.info("launched %d engine replicas (tp=%d each) on '%s'",
                data_parallel_size, tp, queue_name)
    try:
        for proc in procs:
            proc.join()
    except KeyboardInterrupt:
        for proc in procs:
            proc.terminate()
        for proc in procs:
            proc.join(timeout=10)
    finally:
        bad = [proc.exitcode for proc in procs if proc.exitcode not in (0, None)]
        if bad:
            sys.exit(bad[0])


def run_dummy_worker(
    queue_name: str,
    delay_s: float = 1.0,
    prefetch: Optional[int] = None,
    pipeline: Optional[PipelineConfig] = None,
    stage_name: Optional[str] = None,
) -> None:
    setup_logging(json_output=True)
    from llmq_amd.workers.dummy_worker import DummyWorker

    worker = DummyWorker(
        queue_name, delay_s=delay_s, prefetch=prefetch, pipeline=pipeline,
        stage_name=stage_name,
    )
    asyncio.run(worker.run())


def run_semhash_worker(
    queue_name: str,
    mode: str = "dedup",
    batch_size: int = 1000,
    threshold: float = 0.9,
    text_field: Optional[str] = None,
    prefetch: Optional[int] = None,
    pipeline: Optional[PipelineConfig] = None,
    stage_name: Optional[str] = None,
) -> None:
    setup_logging(json_output=True)
    from llmq_amd.workers.semhash_worker import SemHashWorker

    worker = SemHashWorker(
        queue_name,
        mode=mode,
        batch_size=batch_size,
        threshold=threshold,
        text_field=text_field,
        prefetch=prefetch,
        pipeline=pipeline,
        stage_name=stage_name,
    )
    asyncio.run(worker.run())


def run_filter_worker(
    queue_name: str,
    field: str,
    value: str,
    prefetch: Optional[int] = None,
    pipeline: Optional[PipelineConfig] = None,
    stage_name: Optional[str] = None,
) -> None:
    setup_logging(json_output=True)
    from llmq_amd.workers.filter_worker import FieldFilterWorker

    worker = FieldFilterWorker(
        queue_name, field=field, value=value, prefetch=prefetch,
        pipeline=pipeline, stage_name=stage_name,
    )
    asyncio.run(worker.run())


def run_pipeline_worker(
    pipeline_path: str,
    stage_name: str,
    tensor_parallel_size: Optional[int] = None,
    prefetch: Optional[int] = None,
) -> None:
    """Dispatch on the stage's worker type (reference: cli/worker.py:130-250;
    including the 'filter' type the reference documents but never implemented,
    README.md:250 / SURVEY §2 quirks — here it maps to semhash modes)."""
    pipeline = PipelineConfig.from_yaml_file(pipeline_path)
    stage = pipeline.get_stage(stage_name)
    queue_name = pipeline.get_stage_queue_name(stage_name)
    cfg = stage.config or {}

    if stage.worker in ("engine", "vllm"):  # "vllm" accepted for compat
        model = cfg.get("model")
        if not model:
            print(
                f"Stage '{stage_name}' uses the engine worker but has no model configured",
                file=sys.stderr,
            )
            sys.exit(1)
        run_engine_worker(
            model,
            queue_name,
            tensor_parallel_size=tensor_parallel_size or cfg.get("tensor_parallel_size"),
            max_num_seqs=cfg.get("max_num_seqs"),
            max_model_len=cfg.get("max_model_len"),
            prefetch=prefetch,
            pipeline=pipeline,
            stage_name=stage_name,
            stage_config=cfg,
        )
    elif stage.worker == "dummy":
        run_dummy_worker(
            queue_name,
            delay_s=float(cfg.get("delay_s", 1.0)),
            prefetch=prefetch,
            pipeline=pipeline,
            stage_name=stage_name,
        )
    elif stage.worker == "filter" and cfg.get("field") is not None:
        run_filter_worker(
            queue_name,
            field=str(cfg["field"]),
            value=str(cfg.get("value", "")),
            prefetch=prefetch,
            pipeline=pipeline,
            stage_name=stage_name,
        )
    elif stage.worker in ("semhash", "filter"):
        run_semhash_worker(
            queue_name,
            mode=cfg.get("mode", "dedup"),
            batch_size=int(cfg.get("batch_size", 1000)),
            threshold=float(cfg.get("threshold", 0.9)),
            text_field=cfg.get("text_field"),
            prefetch=prefetch,
            pipeline=pipeline,
            stage_name=stage_name,
        )
    else:
        print(f"Unknown worker type '{stage.worker}' for stage '{stage_name}'", file=sys.stderr)
        sys.exit(1)
