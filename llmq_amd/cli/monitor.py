"""Monitoring commands: status / health / errors / clear / pipeline view.

Reference parity: llmq/cli/monitor.py:104-591. Stats come from the in-tree
broker directly (the reference scrapes RabbitMQ's management HTTP API,
broker.py:244-289). Health additionally uses the broker's worker registry
(heartbeats) — the reference infers health from consumer counts only.
"""

from __future__ import annotations

import asyncio
import time
from datetime import datetime, timezone
from typing import List, Optional

from rich.console import Console
from rich.panel import Panel
from rich.table import Table

from llmq_amd.core.client import BrokerClient
from llmq_amd.core.config import get_config
from llmq_amd.core.models import QueueStats
from llmq_amd.core.pipeline import PipelineConfig

console = Console(stderr=True)

BACKLOG_WARN = 1000
BACKLOG_UNHEALTHY = 10000


def _fmt_bytes(n: Optional[int]) -> str:
    if n is None:
        return "-"
    for unit in ("B", "KB", "MB", "GB"):
        if n < 1024:
            return f"{n:.0f}{unit}"
        n /= 1024
    return f"{n:.1f}TB"


def _stats_table(stats_list: List[QueueStats], title: str) -> Table:
    table = Table(title=title)
    table.add_column("queue", style="cyan")
    table.add_column("ready", justify="right")
    table.add_column("unacked", justify="right")
    table.add_column("total", justify="right")
    table.add_column("bytes", justify="right")
    table.add_column("consumers", justify="right")
    table.add_column("source", style="dim")
    for s in stats_list:
        table.add_row(
            s.queue_name,
            str(s.message_count_ready if s.message_count_ready is not None else "-"),
            str(
                s.message_count_unacknowledged
                if s.message_count_unacknowledged is not None
                else "-"
            ),
            str(s.message_count if s.message_count is not None else "-"),
            _fmt_bytes(s.message_bytes),
            str(s.consumer_count if s.consumer_count is not None else "-"),
            s.stats_source,
        )
    return table


async def _connect() -> BrokerClient:
    client = BrokerClient(get_config())
    await client.connect(retries=1)
    return client


def show_status(queue_name: Optional[str]) -> None:
    async def inner() -> None:
        try:
            client = await _connect()
        except ConnectionError as exc:
            console.print(f"[red]Cannot reach broker:[/red] {exc}")
            raise SystemExit(1)
        if queue_name:
            stats = [await client.get_queue_stats(queue_name)]
            for suffix in (".results", ".failed"):
                extra = await client.get_queue_stats(queue_name + suffix)
                if extra.stats_source != "unavailable":
                    stats.append(extra)
        else:
            stats = await client.list_queues()
        console.print(_stats_table(stats, "Queue status"))
        await client.disconnect()

    asyncio.run(inner())


def show_connection_status() -> None:
    async def inner() -> None:
        config = get_config()
        try:
            client = await _connect()
            reply = await client.call({"m": "ping"})
            console.print(
                Panel(
                    f"[green]Connected[/green] to {config.broker_url}\n"
                    f"uptime: {reply.get('uptime', 0):.0f}s",
                    title="broker",
                )
            )
            await client.disconnect()
        except ConnectionError as exc:
            console.print(Panel(f"[red]Unreachable:[/red] {exc}", title="broker"))
            raise SystemExit(1)

    asyncio.run(inner())


def check_health(queue_name: Optional[str]) -> None:
    async def inner() -> None:
        try:
            client = await _connect()
        except ConnectionError as exc:
            console.print(f"[red]Cannot reach broker:[/red] {exc}")
            raise SystemExit(1)
        workers = await client.get_workers()
        now = time.time()
        table = Table(title="Workers")
        table.add_column("worker", style="cyan")
        table.add_column("queue")
        table.add_column("status")
        table.add_column("last seen", justify="right")
        table.add_column("jobs", justify="right")
        table.add_column("avg ms", justify="right")
        shown = 0
        for w in workers:
            if queue_name and w.get("queue") != queue_name:
                continue
            age = now - w["last_seen"]
            status = w["status"]
            if status == "active" and age > 30:
                status = "stale"
            color = {"active": "green", "stale": "yellow", "stopped": "dim"}.get(status, "white")
            table.add_row(
                w["worker_id"],
                w.get("queue", ""),
                f"[{color}]{status}[/{color}]",
                f"{age:.0f}s ago",
                str(w.get("jobs_processed", 0)),
                f"{w['avg_duration_ms']:.1f}" if w.get("avg_duration_ms") else "-",
            )
            shown += 1
        if shown:
            console.print(table)
        # Queue-level health (reference heuristic: consumers>0, backlog<10000,
        # monitor.py:57-70)
        if queue_name:
            stats = await client.get_queue_stats(queue_name)
            healthy = True
            problems = []
            if not stats.consumer_count:
                healthy = False
                problems.append("no consumers")
            if (stats.message_count_ready or 0) > BACKLOG_UNHEALTHY:
                healthy = False
                problems.append(f"backlog {stats.message_count_ready} > {BACKLOG_UNHEALTHY}")
            verdict = "[green]HEALTHY[/green]" if healthy else "[red]UNHEALTHY[/red]"
            console.print(
                Panel(f"{verdict}" + (f" — {', '.join(problems)}" if problems else ""),
                      title=f"queue {queue_name}")
            )
            if not healthy:
                raise SystemExit(1)
        elif not shown:
            console.print("[yellow]No workers have reported in.[/yellow]")
        await client.disconnect()

    asyncio.run(inner())


def show_errors(queue_name: str, limit: int) -> None:
    async def inner() -> None:
        try:
            client = await _connect()
        except ConnectionError as exc:
            console.print(f"[red]Cannot reach broker:[/red] {exc}")
            raise SystemExit(1)
        errors = await client.get_failed_messages(queue_name, limit)
        if not errors:
            console.print(f"[green]No failed jobs on '{queue_name}.failed'[/green]")
        else:
            table = Table(title=f"Failed jobs — {queue_name}.failed (showing {len(errors)})")
            table.add_column("job id", style="cyan")
            table.add_column("error")
            table.add_column("worker", style="dim")
            table.add_column("when", style="dim")
            for e in errors:
                ts = e.timestamp
                if isinstance(ts, datetime):
                    when = ts.astimezone(timezone.utc).strftime("%H:%M:%S")
                else:
                    when = str(ts)
                table.add_row(e.job_id, e.error_message[:80], e.worker_id or "-", when)
            console.print(table)
        await client.disconnect()

    asyncio.run(inner())


def clear_queue(queue_name: str, include_results: bool, yes: bool) -> None:
    async def inner() -> None:
        try:
            client = await _connect()
        except ConnectionError as exc:
            console.print(f"[red]Cannot reach broker:[/red] {exc}")
            raise SystemExit(1)
        targets = [queue_name]
        if include_results:
            targets += [queue_name + ".results", queue_name + ".failed"]
        for t in targets:
            purged = await client.clear_queue(t)
            console.print(f"Purged {purged} messages from '{t}'")
        await client.disconnect()

    if not yes:
        confirm = input(f"Clear queue '{queue_name}'? [y/N] ")
        if confirm.strip().lower() not in ("y", "yes"):
            console.print("aborted")
            return
    asyncio.run(inner())


def show_pipeline_status(pipeline_path: str) -> None:
    async def inner() -> None:
        pipeline = PipelineConfig.from_yaml_file(pipeline_path)
        try:
            client = await _connect()
        except ConnectionError as exc:
            console.print(f"[red]Cannot reach broker:[/red] {exc}")
            raise SystemExit(1)
        stats = []
        warnings = []
        for stage in pipeline.stages:
            qname = pipeline.get_stage_queue_name(stage.name)
            s = await client.get_queue_stats(qname)
            stats.append(s)
            if not s.consumer_count:
                warnings.append(f"stage '{stage.name}': NO WORKERS")
            if (s.message_count_ready or 0) > BACKLOG_WARN:
                warnings.append(
                    f"stage '{stage.name}': backlog {s.message_count_ready} > {BACKLOG_WARN}"
                )
        results_stats = await client.get_queue_stats(
            pipeline.get_pipeline_results_queue_name()
        )
        stats.append(results_stats)
        console.print(_stats_table(stats, f"Pipeline '{pipeline.name}'"))
        flow = " → ".join(
            f"{s.name}[{st.message_count or 0}]" for s, st in zip(pipeline.stages, stats)
        )
        console.print(Panel(flow + f" → results[{results_stats.message_count or 0}]",
                            title="flow"))
        for w in warnings:
            console.print(f"[yellow]⚠ {w}[/yellow]")
        await client.disconnect()

    asyncio.run(inner())
