"""CLI entry point (reference parity: llmq/cli/main.py:6-546).

Command tree:
  llmq broker serve            start the in-tree broker (replaces RabbitMQ)
  llmq submit <queue> <src>    submit jobs (JSONL / stdin / dataset)
  llmq receive <queue>         drain results to stdout JSONL
  llmq status [queue]          queue stats
  llmq health [queue]          worker + queue health
  llmq errors <queue>          dead-letter queue view
  llmq clear <queue>           purge a queue
  llmq worker run|dummy|semhash|pipeline
Operation modules are imported lazily (reference main.py:102,290-294) so
`llmq --help` stays fast and CPU-only commands never import torch.
"""

from __future__ import annotations

import sys
from typing import Tuple

import click

from llmq_amd.utils.logging import setup_logging


def _parse_map(map_args: Tuple[str, ...]) -> dict:
    mapping = {}
    for item in map_args:
        if "=" not in item:
            raise click.BadParameter(f"--map expects var=column, got '{item}'")
        var, _, col = item.partition("=")
        mapping[var.strip()] = col.strip()
    return mapping


@click.group()
@click.version_option(package_name=None, version="0.1.0", prog_name="llmq")
def cli() -> None:
    """llmq-amd — MI355X-native distributed batch inference."""


# ---------------------------------------------------------------- broker --


@cli.group()
def broker() -> None:
    """In-tree message broker."""


@broker.command("serve")
@click.option("--host", default=None, help="Bind host (default from LLMQ_BROKER_URL)")
@click.option("--port", type=int, default=None, help="Bind port")
@click.option("--data-dir", default=None, help="Durable spool directory")
@click.option("--max-retries", type=int, default=None)
@click.option("--ephemeral", is_flag=True, help="No durable spool (tests)")
def broker_serve(host, port, data_dir, max_retries, ephemeral) -> None:
    """Run the broker until interrupted."""
    setup_logging()
    import asyncio

    from llmq_amd.broker.server import run_broker
    from llmq_amd.core.config import get_config

    config = get_config()
    host = host or config.broker_host
    port = port if port is not None else config.broker_port
    data_dir = None if ephemeral else (data_dir or config.broker_data_dir)
    retries = max_retries if max_retries is not None else config.max_retries
    try:
        asyncio.run(run_broker(host, port, data_dir, retries))
    except KeyboardInterrupt:
        pass


# ---------------------------------------------------------------- submit --


@cli.command()
@click.argument("queue_name")
@click.argument("source")
@click.option("--template", default=None, help="Prompt template with {placeholders}")
@click.option("--map", "map_args", multiple=True, help="var=column mapping")
@click.option("--limit", type=int, default=None, help="Max jobs to submit")
@click.option("--max-samples", type=int, default=None,
              help="Alias for --limit (reference main.py:40-42)")
@click.option("--split", default="train", help="Dataset split (default: train)")
@click.option("--subset", default=None, help="Dataset subset/config name")
@click.option("--timeout", "stream_timeout", type=float, default=300.0,
              help="Idle timeout in seconds waiting for results (with --stream)")
@click.option("--stream", is_flag=True, help="Echo results to stdout as they arrive")
@click.option("-p", "--pipeline", "pipeline_path", default=None,
              help="Pipeline YAML: submit to its first stage")
def submit(queue_name, source, template, map_args, limit, max_samples, split,
           subset, stream_timeout, stream, pipeline_path) -> None:
    """Submit jobs from SOURCE (jsonl path, '-' for stdin, or dataset id).

    With -p/--pipeline, QUEUE_NAME is ignored and jobs go to the pipeline's
    first stage queue."""
    setup_logging()
    from llmq_amd.cli.submit import run_pipeline_submit, run_submit

    mapping = _parse_map(map_args)
    if limit is None:
        limit = max_samples
    if pipeline_path:
        run_pipeline_submit(pipeline_path, source, mapping, limit, stream,
                            split=split, subset=subset,
                            stream_timeout=stream_timeout)
    else:
        run_submit(queue_name, source, template, mapping, limit, stream,
                   split=split, subset=subset, stream_timeout=stream_timeout)


@cli.command("pipeline", deprecated=True)
@click.argument("pipeline_config_path")
@click.argument("jobs_source")
@click.option("--timeout", "stream_timeout", type=float, default=300.0)
@click.option("--map", "map_args", multiple=True)
@click.option("--max-samples", type=int, default=None)
@click.option("--split", default="train")
@click.option("--subset", default=None)
@click.option("--stream", is_flag=True)
def pipeline_submit_deprecated(pipeline_config_path, jobs_source, stream_timeout,
                               map_args, max_samples, split, subset, stream):
    """[deprecated] Use `submit -p PIPELINE.yaml` (reference main.py:150-175
    keeps this alias; kept for command-line compatibility)."""
    setup_logging()
    from llmq_amd.cli.submit import run_pipeline_submit

    run_pipeline_submit(pipeline_config_path, jobs_source, _parse_map(map_args),
                        max_samples, stream, split=split, subset=subset,
                        stream_timeout=stream_timeout)


@cli.command("receive-pipeline", deprecated=True)
@click.argument("pipeline_config_path")
@click.option("--timeout", type=float, default=300.0)
def receive_pipeline_deprecated(pipeline_config_path, timeout):
    """[deprecated] Use `receive -p PIPELINE.yaml` (reference main.py:376-378)."""
    setup_logging()
    from llmq_amd.cli.receive import run_pipeline_receive

    run_pipeline_receive(pipeline_config_path, timeout, None)


# --------------------------------------------------------------- receive --


@cli.command()
@click.argument("queue_name", required=False)
@click.option("--timeout", type=float, default=300.0,
              help="Idle timeout seconds (reference default 300, main.py:334)")
@click.option("--limit", type=int, default=None)
@click.option("--skip-filtered", is_flag=True, help="Drop filtered (semhash) results")
@click.option("-p", "--pipeline", "pipeline_path", default=None,
              help="Pipeline YAML: receive its final results")
def receive(queue_name, timeout, limit, skip_filtered, pipeline_path) -> None:
    """Drain results of QUEUE_NAME (or a pipeline) to stdout as JSONL."""
    setup_logging()
    from llmq_amd.cli.receive import run_pipeline_receive, run_receive

    if pipeline_path:
        run_pipeline_receive(pipeline_path, timeout, limit)
    elif queue_name:
        run_receive(queue_name, timeout, limit, skip_filtered)
    else:
        raise click.UsageError("Provide a queue name or -p pipeline.yaml")


# ---------------------------------------------------------------- status --


@cli.command()
@click.argument("queue_name", required=False)
@click.option("-p", "--pipeline", "pipeline_path", default=None)
def status(queue_name, pipeline_path) -> None:
    """Queue depths / consumers (all queues if none given)."""
    setup_logging()
    from llmq_amd.cli import monitor

    if pipeline_path:
        monitor.show_pipeline_status(pipeline_path)
    elif queue_name:
        monitor.show_status(queue_name)
    else:
        monitor.show_status(None)


@cli.command()
@click.argument("queue_name", required=False)
def health(queue_name) -> None:
    """Worker heartbeats + queue health verdict."""
    setup_logging()
    from llmq_amd.cli import monitor

    monitor.check_health(queue_name)


@cli.command()
@click.argument("queue_name")
@click.option("--limit", type=int, default=100,
              help="Max errors to show (reference default, main.py:318)")
def errors(queue_name, limit) -> None:
    """Show dead-lettered jobs from <queue>.failed."""
    setup_logging()
    from llmq_amd.cli import monitor

    monitor.show_errors(queue_name, limit)


@cli.command()
@click.argument("queue_name")
@click.option("--include-results", is_flag=True, help="Also purge .results and .failed")
@click.option("-y", "--yes", is_flag=True)
def clear(queue_name, include_results, yes) -> None:
    """Purge all ready messages from a queue."""
    setup_logging()
    from llmq_amd.cli import monitor

    monitor.clear_queue(queue_name, include_results, yes)


# ---------------------------------------------------------------- worker --


@cli.group()
def worker() -> None:
    """Run workers."""


@worker.command("run")
@click.argument("model")
@click.argument("queue_name")
@click.option("--tensor-parallel-size", "-tp", type=int, default=None,
              help="GPUs to shard the model over (RCCL/xGMI)")
@click.option("--max-num-seqs", type=int, default=None)
@click.option("--max-model-len", type=int, default=None)
@click.option("--prefetch", type=int, default=None)
@click.option("--kv-cache-dtype", type=click.Choice(["auto", "fp8"]), default="auto",
              help="KV storage dtype; fp8 (OCP e4m3) doubles KV capacity")
@click.option("--engine-overrides", default=None,
              help='JSON dict of EngineConfig overrides, e.g. \'{"enforce_eager": true}\'')
@click.option("--data-parallel-size", "-dp", type=int, default=None,
              help="Model replicas to launch, one process per GPU set "
                   "(reference main.py:433-439 passes this to vLLM; here it "
                   "spawns N workers competing on the same queue)")
def worker_run(model, queue_name, tensor_parallel_size, max_num_seqs, max_model_len,
               prefetch, kv_cache_dtype, engine_overrides, data_parallel_size):
    """GPU inference worker (in-tree MI355X engine)."""
    import json as _json

    from llmq_amd.cli.worker import run_engine_worker, run_engine_worker_dp

    overrides = _json.loads(engine_overrides) if engine_overrides else {}
    if kv_cache_dtype != "auto":
        overrides.setdefault("kv_cache_dtype", kv_cache_dtype)
    if data_parallel_size and data_parallel_size > 1:
        run_engine_worker_dp(
            model, queue_name, data_parallel_size,
            tensor_parallel_size=tensor_parallel_size,
            max_num_seqs=max_num_seqs,
            max_model_len=max_model_len,
            prefetch=prefetch,
            engine_overrides=overrides or None,
        )
        return
    run_engine_worker(
        model, queue_name,
        tensor_parallel_size=tensor_parallel_size,
        max_num_seqs=max_num_seqs,
        max_model_len=max_model_len,
        prefetch=prefetch,
        engine_overrides=overrides or None,
    )


@worker.command("dummy")
@click.argument("queue_name")
@click.option("--delay", type=float, default=1.0, help="Seconds per job")
@click.option("--prefetch", type=int, default=None)
@click.option("--concurrency", "-c", type=int, default=None,
              help="In-flight jobs (reference main.py:466-472; maps to prefetch)")
def worker_dummy(queue_name, delay, prefetch, concurrency):
    """CPU echo worker (tests/plumbing)."""
    from llmq_amd.cli.worker import run_dummy_worker

    run_dummy_worker(queue_name, delay_s=delay, prefetch=prefetch or concurrency)


@worker.command("semhash")
@click.argument("queue_name")
@click.option("--mode", type=click.Choice([
                  "dedup", "outliers", "representatives",
                  # reference mode names (main.py:487-492) accepted as aliases
                  "deduplicate", "filter_outliers", "find_representative"]),
              default="dedup")
@click.option("--batch-size", type=int, default=1000)
@click.option("--threshold", type=float, default=0.9)
@click.option("--text-field", default=None)
@click.option("--prefetch", type=int, default=None)
@click.option("--concurrency", "-c", type=int, default=None,
              help="In-flight jobs (reference alias; maps to prefetch)")
def worker_semhash(queue_name, mode, batch_size, threshold, text_field, prefetch,
                   concurrency):
    """Semantic dedup / filter worker."""
    from llmq_amd.cli.worker import run_semhash_worker

    mode = {"deduplicate": "dedup", "filter_outliers": "outliers",
            "find_representative": "representatives"}.get(mode, mode)
    run_semhash_worker(
        queue_name, mode=mode, batch_size=batch_size, threshold=threshold,
        text_field=text_field, prefetch=prefetch or concurrency,
    )


@worker.command("filter")
@click.argument("queue_name")
@click.argument("field")
@click.argument("value")
@click.option("--prefetch", type=int, default=None)
def worker_filter(queue_name, field, value, prefetch):
    """Pass jobs whose FIELD equals VALUE; mark the rest filtered
    (reference README.md:250 documents this worker but never implemented it)."""
    from llmq_amd.cli.worker import run_filter_worker

    run_filter_worker(queue_name, field, value, prefetch=prefetch)


@worker.command("pipeline")
@click.argument("pipeline_path")
@click.argument("stage_name")
@click.option("--tensor-parallel-size", "-tp", type=int, default=None)
@click.option("--prefetch", type=int, default=None)
def worker_pipeline(pipeline_path, stage_name, tensor_parallel_size, prefetch):
    """Run the worker for one pipeline stage."""
    from llmq_amd.cli.worker import run_pipeline_worker

    run_pipeline_worker(pipeline_path, stage_name, tensor_parallel_size, prefetch)


def main() -> None:
    try:
        cli()
    except BrokenPipeError:
        sys.exit(0)


if __name__ == "__main__":
    main()
