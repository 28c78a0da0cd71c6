"""Job submission (reference parity: llmq/cli/submit.py:28-941).

Input sources (sniffed like the reference, submit.py:78-94):
  "-"                 → stdin JSONL
  existing file path  → JSONL file
  anything else with "/" or no .jsonl/.json/.txt suffix → HF dataset id
                        (loaded via `datasets` if importable/local)

Rows without prompt/messages are wrapped with --template / --map column
mapping. Publishes in chunks (LLMQ_CHUNK_SIZE, default 10000) via the
broker's batch publish. ``--stream`` echoes results to stdout as they
arrive.
"""

from __future__ import annotations

import asyncio
import json
import logging
import signal
import sys
import time
import uuid
from pathlib import Path
from typing import Any, Dict, Iterator, List, Optional

from llmq_amd.core.client import BrokerClient, Delivery
from llmq_amd.core.config import get_config
from llmq_amd.core.models import Job, Result
from llmq_amd.core.pipeline import PipelineConfig
from llmq_amd.utils.template import create_job_from_data

logger = logging.getLogger(__name__)


class SubmitProgress:
    """Rich progress with live submit/complete rates (reference parity:
    submit.py:350-364,437-448). Renders on stderr ONLY when it is a
    terminal, so ``--stream``'s JSONL on stdout and piped/captured stderr
    stay clean; otherwise falls back to periodic plain lines."""

    def __init__(self, stream: bool, total: Optional[int] = None):
        from rich.console import Console

        self._console = Console(file=sys.stderr)
        self.enabled = self._console.is_terminal
        self._progress = None
        self._submit_task = None
        self._recv_task = None
        self.stream = stream
        self.total = total

    def __enter__(self) -> "SubmitProgress":
        if not self.enabled:
            return self
        from rich.progress import (
            BarColumn, MofNCompleteColumn, Progress, ProgressColumn,
            SpinnerColumn, TextColumn, TimeElapsedColumn,
        )
        from rich.text import Text

        class RateColumn(ProgressColumn):
            def render(self, task):  # noqa: ANN001
                speed = task.speed
                return Text(f"{speed:.1f} jobs/s" if speed else "-- jobs/s",
                            style="progress.data.speed")

        self._progress = Progress(
            SpinnerColumn(), TextColumn("[progress.description]{task.description}"),
            BarColumn(), MofNCompleteColumn(), RateColumn(), TimeElapsedColumn(),
            console=self._console, transient=False,
            # rich's Live redirects stdout into the progress console by
            # default — that would swallow --stream's result JSONL. Keep
            # both streams untouched; the bar owns only its own console.
            redirect_stdout=False, redirect_stderr=False,
        )
        self._progress.__enter__()
        self._submit_task = self._progress.add_task("submit", total=self.total)
        if self.stream:
            self._recv_task = self._progress.add_task("results", total=self.total)
        return self

    def __exit__(self, *exc) -> None:
        if self._progress is not None:
            self._progress.__exit__(*exc)

    def update_submitted(self, n: int) -> None:
        if self._progress is not None:
            self._progress.update(self._submit_task, completed=n)
        else:
            print(f"  submitted {n} jobs...", file=sys.stderr, end="\r")

    def update_received(self, n: int, total: Optional[int] = None) -> None:
        if self._progress is not None and self._recv_task is not None:
            self._progress.update(self._recv_task, completed=n, total=total)


def _looks_like_dataset(source: str) -> bool:
    if source == "-":
        return False
    p = Path(source)
    if p.exists():
        # an existing directory is a local dataset; an existing file is JSONL
        return p.is_dir()
    return "/" in source or p.suffix not in (".jsonl", ".json", ".txt")


def _iter_file_rows(source: str) -> Iterator[Dict[str, Any]]:
    fh = sys.stdin if source == "-" else open(source, "r", encoding="utf-8")
    try:
        for lineno, line in enumerate(fh, 1):
            line = line.strip()
            if not line:
                continue
            try:
                yield json.loads(line)
            except json.JSONDecodeError as exc:
                logger.warning("skipping line %d: %s", lineno, exc)
    finally:
        if source != "-":
            fh.close()


def _iter_dataset_rows(
    source: str, split: str = "train", subset: Optional[str] = None
) -> Iterator[Dict[str, Any]]:
    try:
        import datasets  # noqa: PLC0415
    except ImportError as exc:  # pragma: no cover
        raise RuntimeError("`datasets` not available for dataset sources") from exc
    ds = datasets.load_dataset(source, name=subset, split=split, streaming=True)
    for row in ds:
        yield dict(row)


class JobSubmitter:
    def __init__(
        self,
        queue_name: str,
        source: str,
        template: Optional[str] = None,
        column_mapping: Optional[Dict[str, str]] = None,
        limit: Optional[int] = None,
        stream: bool = False,
        split: str = "train",
        subset: Optional[str] = None,
        stream_timeout: float = 300.0,
        id_prefix: Optional[str] = None,
    ):
        self.queue_name = queue_name
        self.source = source
        self.template = template
        self.column_mapping = column_mapping or {}
        self.limit = limit
        self.stream = stream
        self.split = split
        self.subset = subset
        self.stream_timeout = stream_timeout
        self.id_prefix = id_prefix or ("dataset" if _looks_like_dataset(source) else "job")
        self.config = get_config()
        self.client = BrokerClient(self.config)
        self.submitted = 0
        self.received = 0
        self._progress = SubmitProgress(stream=False)  # replaced in run()
        self._interrupted = 0

    def _rows(self) -> Iterator[Dict[str, Any]]:
        if _looks_like_dataset(self.source):
            return _iter_dataset_rows(self.source, self.split, self.subset)
        return _iter_file_rows(self.source)

    def _make_job(self, idx: int, row: Dict[str, Any]) -> Job:
        job_id = row.get("id") or f"{self.id_prefix}-{idx:08d}-{uuid.uuid4().hex[:8]}"
        return create_job_from_data(row, job_id, self.template, self.column_mapping)

    def _handle_sigint(self, *_a: Any) -> None:
        self._interrupted += 1
        if self._interrupted >= 2:
            sys.exit(130)

    async def run(self) -> int:
        loop = asyncio.get_event_loop()
        try:
            loop.add_signal_handler(signal.SIGINT, self._handle_sigint)
        except (NotImplementedError, RuntimeError):
            pass
        await self.client.connect()
        await self.client.setup_queue_infrastructure(self.queue_name)
        stream_task = None
        expected_ids: set[str] = set()
        if self.stream:
            stream_task = asyncio.create_task(
                self._consume_results(expected_ids, idle_timeout=self.stream_timeout))
        start = time.time()
        chunk: List[Job] = []
        skipped = 0
        self._progress = SubmitProgress(self.stream, total=self.limit)
        with self._progress:
            for idx, row in enumerate(self._rows()):
                if self._interrupted:
                    break
                if self.limit is not None and self.submitted + len(chunk) >= self.limit:
                    break
                try:
                    job = self._make_job(idx, row)
                except (ValueError, KeyError) as exc:
                    skipped += 1
                    logger.warning("skipping row %d: %s", idx, exc)
                    continue
                chunk.append(job)
                if self.stream:
                    expected_ids.add(job.id)
                if len(chunk) >= self.config.chunk_size:
                    await self._submit_chunk(chunk)
                    chunk = []
            if chunk and not self._interrupted:
                await self._submit_chunk(chunk)
            if stream_task is not None:
                self._progress.update_received(self.received, total=self.submitted)
                await stream_task
        elapsed = max(time.time() - start, 1e-9)
        print(
            f"Submitted {self.submitted} jobs to '{self.queue_name}' "
            f"in {elapsed:.1f}s ({self.submitted / elapsed:.1f} jobs/s)"
            + (f", skipped {skipped}" if skipped else ""),
            file=sys.stderr,
        )
        if stream_task is not None:
            print(
                f"Received {self.received}/{self.submitted} results", file=sys.stderr
            )
        else:
            print(
                f"Results will accumulate on '{self.queue_name}.results' — "
                f"run `llmq receive {self.queue_name}` to collect them.",
                file=sys.stderr,
            )
        await self.client.disconnect()
        return self.submitted

    async def _submit_chunk(self, jobs: List[Job]) -> None:
        await self.client.publish_jobs(self.queue_name, jobs)
        self.submitted += len(jobs)
        self._progress.update_submitted(self.submitted)

    async def _consume_results(self, expected_ids: set[str], idle_timeout: float = 60.0) -> None:
        done = asyncio.Event()
        last_seen = time.time()

        async def on_result(delivery: Delivery) -> None:
            nonlocal last_seen
            last_seen = time.time()
            try:
                result = Result.model_validate_json(delivery.body)
            except Exception:
                await delivery.nack(requeue=False, error="bad result")
                return
            if result.id in expected_ids:
                expected_ids.discard(result.id)
                print(delivery.body, flush=True)
                self.received += 1
                self._progress.update_received(
                    self.received, total=self.submitted or None)
                await delivery.ack()
                if not expected_ids and self.submitted:
                    done.set()
            else:
                await delivery.nack(requeue=True)

        await self.client.consume_results(self.queue_name, on_result, prefetch=1000)
        while not done.is_set() and not self._interrupted:
            try:
                await asyncio.wait_for(done.wait(), timeout=1.0)
            except asyncio.TimeoutError:
                if time.time() - last_seen > idle_timeout and self.submitted:
                    logger.warning("idle timeout waiting for results")
                    break


class PipelineSubmitter(JobSubmitter):
    def __init__(self, pipeline: PipelineConfig, source: str, **kwargs):
        self.pipeline = pipeline
        first_stage = pipeline.stages[0]
        template = kwargs.pop("template", None) or first_stage.template
        if template is None and first_stage.messages is not None:
            # chat-template first stage (reference example-pipeline.yaml):
            # route rows through the JSON --map machinery so {vars}
            # interpolate into the messages list per row
            import json as _json

            mapping = dict(kwargs.pop("column_mapping", None) or {})
            mapping.setdefault("messages", _json.dumps(first_stage.messages))
            kwargs["column_mapping"] = mapping
        super().__init__(
            pipeline.get_stage_queue_name(first_stage.name),
            source,
            template=template,
            **kwargs,
        )

    async def run(self) -> int:
        await self.client.connect()
        await self.client.setup_pipeline_infrastructure(self.pipeline)
        await self.client.disconnect()
        return await super().run()

    async def _consume_results(self, expected_ids: set[str], idle_timeout: float = 60.0) -> None:
        # Final results live on pipeline.<name>.results
        self.queue_name = self.pipeline.get_pipeline_results_queue_name().removesuffix(
            ".results"
        )
        await super()._consume_results(expected_ids, idle_timeout)


def run_submit(
    queue_name: str,
    source: str,
    template: Optional[str],
    column_mapping: Dict[str, str],
    limit: Optional[int],
    stream: bool,
    split: str = "train",
    subset: Optional[str] = None,
    stream_timeout: float = 300.0,
) -> None:
    submitter = JobSubmitter(
        queue_name, source, template=template, column_mapping=column_mapping,
        limit=limit, stream=stream, split=split, subset=subset,
        stream_timeout=stream_timeout,
    )
    asyncio.run(submitter.run())


def run_pipeline_submit(
    pipeline_path: str,
    source: str,
    column_mapping: Dict[str, str],
    limit: Optional[int],
    stream: bool,
    split: str = "train",
    subset: Optional[str] = None,
    stream_timeout: float = 300.0,
) -> None:
    pipeline = PipelineConfig.from_yaml_file(pipeline_path)
    submitter = PipelineSubmitter(
        pipeline, source, column_mapping=column_mapping, limit=limit,
        stream=stream, split=split, subset=subset, stream_timeout=stream_timeout,
    )
    asyncio.run(submitter.run())
