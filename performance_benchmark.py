#!/usr/bin/env python3
"""End-to-end serving benchmark through the full queue stack.

The MI355X counterpart of the reference harness
(/root/reference/performance_benchmark.py:100-693): orchestrate broker +
worker subprocess(es) + submit + receive as real processes, sweep the
engine batch cap (max_num_seqs — the reference sweeps VLLM_MAX_NUM_SEQS,
performance_benchmark.py:645-667) and report input/output/total
tokens-per-second, jobs/sec and avg/p95/p99 end-to-end latency
(performance_benchmark.py:329-360).

Differences from the reference, by design:
- The broker is the in-tree llmq broker (no external RabbitMQ container).
- Token counts come from the engine's Result.prompt_tokens /
  Result.output_tokens — exact, not tiktoken-estimated (reference
  performance_benchmark.py:91-97 falls back to len/4).
- GPU info via rocm-smi/amd-smi (reference uses nvidia-smi, 117-126).
- Jobs are synthetic (no network for HF datasets here); --dataset accepts
  a JSONL path for real prompts.

Usage:
  python performance_benchmark.py --model tower-plus-9b --samples 2000 \
      --batch-sizes 64,128,256 --max-tokens 128 --output results.json
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import signal
import statistics
import subprocess
import sys
import time
from dataclasses import asdict, dataclass, field
from typing import Any, Dict, List, Optional

REPO_ROOT = os.path.dirname(os.path.abspath(__file__))


@dataclass
class RunResult:
    """One sweep point (reference BenchmarkResult, performance_benchmark.py:48-70)."""

    max_num_seqs: int
    total_requests: int
    successful_requests: int
    failed_requests: int
    total_time_seconds: float
    total_input_tokens: int
    total_output_tokens: int
    input_tokens_per_second: float
    output_tokens_per_second: float
    total_tokens_per_second: float
    jobs_per_second: float
    avg_processing_time_ms: float
    avg_queue_wait_ms: float
    avg_total_latency_ms: float
    p95_total_latency_ms: float
    p99_total_latency_ms: float


def gpu_info() -> Dict[str, Any]:
    """GPU inventory via rocm-smi (reference: nvidia-smi, 117-126)."""
    for cmd in (
        ["rocm-smi", "--showproductname", "--json"],
        ["amd-smi", "list", "--json"],
    ):
        try:
            out = subprocess.run(cmd, capture_output=True, text=True, timeout=20)
            if out.returncode == 0 and out.stdout.strip():
                return {"tool": cmd[0], "raw": json.loads(out.stdout)}
        except (OSError, json.JSONDecodeError, subprocess.TimeoutExpired):
            continue
    return {"tool": None, "raw": {}}


def synthetic_jobs(
    n: int, prompt_tokens: int, long_frac: float = 0.0, long_tokens: int = 0
) -> List[Dict[str, Any]]:
    """Deterministic synthetic prompts of ~prompt_tokens byte-tokens;
    every ⌈1/long_frac⌉-th job gets a long_tokens prompt (mixed workloads —
    exercises chunked prefill + mixed steps)."""
    base = "the quick brown fox jumps over the lazy dog. "

    def body(tokens: int) -> str:
        return (base * (tokens // len(base) + 1))[:tokens]

    stride = int(1 / long_frac) if long_frac > 0 else 0
    out = []
    for i in range(n):
        tokens = long_tokens if (stride and i % stride == 0) else prompt_tokens
        out.append({"id": f"bench-{i:08d}", "prompt": f"[{i}] {body(tokens)}"})
    return out


class Orchestrator:
    def __init__(self, args: argparse.Namespace):
        self.args = args
        self.port = args.port
        self.env = dict(
            os.environ,
            LLMQ_BROKER_URL=f"llmq://127.0.0.1:{self.port}",
            LLMQ_QUEUE_PREFETCH=str(args.prefetch),
        )
        self.procs: List[subprocess.Popen] = []

    # -- process management ---------------------------------------------

    def _spawn(self, argv: List[str], log_path: str, extra_env: Optional[Dict[str, str]] = None) -> subprocess.Popen:
        env = dict(self.env)
        if extra_env:
            env.update(extra_env)
        log = open(log_path, "ab")
        proc = subprocess.Popen(
            argv, stdout=log, stderr=subprocess.STDOUT, env=env, cwd=REPO_ROOT
        )
        self.procs.append(proc)
        return proc

    def start_broker(self) -> None:
        self._spawn(
            [sys.executable, "-m", "llmq_amd", "broker", "serve",
             "--port", str(self.port), "--ephemeral"],
            os.path.join(self.args.log_dir, "broker.log"),
        )
        self._wait_port()

    def _wait_port(self, timeout: float = 30.0) -> None:
        import socket

        deadline = time.time() + timeout
        while time.time() < deadline:
            try:
                with socket.create_connection(("127.0.0.1", self.port), timeout=1):
                    return
            except OSError:
                time.sleep(0.2)
        raise RuntimeError(f"broker did not come up on :{self.port}")

    def start_workers(self, max_num_seqs: int, queue: str) -> List[subprocess.Popen]:
        """One worker process per GPU (reference slurm pattern,
        utils/run_dutch_nemotron.slurm:50-74: HIP_VISIBLE_DEVICES=i)."""
        workers = []
        for i in range(self.args.workers):
            extra = {}
            if self.args.gpus:
                extra["HIP_VISIBLE_DEVICES"] = str(i % self.args.gpus)
            argv = [
                sys.executable, "-m", "llmq_amd", "worker", "run",
                self.args.model, queue,
                "--max-num-seqs", str(max_num_seqs),
            ]
            if self.args.max_model_len:
                argv += ["--max-model-len", str(self.args.max_model_len)]
            if self.args.engine_overrides:
                argv += ["--engine-overrides", self.args.engine_overrides]
            workers.append(
                self._spawn(argv, os.path.join(self.args.log_dir, f"worker{i}.log"), extra)
            )
        return workers

    def wait_worker_ready(self, timeout: float = 600.0) -> None:
        """Grep worker logs for the consume banner (reference waits for
        'starting to consume from queue', performance_benchmark.py:506-510)."""
        deadline = time.time() + timeout
        paths = [
            os.path.join(self.args.log_dir, f"worker{i}.log")
            for i in range(self.args.workers)
        ]
        ready: set = set()
        while time.time() < deadline and len(ready) < len(paths):
            for p in paths:
                if p in ready or not os.path.exists(p):
                    continue
                with open(p, "rb") as fh:
                    if b"starting to consume from queue" in fh.read():
                        ready.add(p)
            # fail FAST if a worker process already died (e.g. KV budget
            # misconfiguration) instead of burning the full timeout
            for proc in self.procs:
                if proc.poll() not in (None, 0):
                    raise RuntimeError(
                        f"worker exited with {proc.returncode} before ready "
                        f"(see {self.args.log_dir}/worker*.log)"
                    )
            time.sleep(0.5)
        if len(ready) < len(paths):
            raise RuntimeError(f"only {len(ready)}/{len(paths)} workers became ready")

    def stop_all(self) -> None:
        for p in self.procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        deadline = time.time() + 20
        for p in self.procs:
            try:
                p.wait(timeout=max(0.1, deadline - time.time()))
            except subprocess.TimeoutExpired:
                p.kill()
        self.procs.clear()

    # -- measurement ----------------------------------------------------

    async def run_point(self, max_num_seqs: int) -> RunResult:
        from llmq_amd.core.client import BrokerClient
        from llmq_amd.core.config import get_config
        from llmq_amd.core.models import Job, Result

        queue = f"bench-{max_num_seqs}"
        os.environ["LLMQ_BROKER_URL"] = f"llmq://127.0.0.1:{self.port}"
        config = get_config()
        client = BrokerClient(config)
        await client.connect()
        await client.setup_queue_infrastructure(queue)

        workers = self.start_workers(max_num_seqs, queue)
        self.wait_worker_ready()

        jobs_data = synthetic_jobs(
            self.args.samples, self.args.prompt_tokens,
            self.args.long_frac, self.args.long_prompt_tokens,
        )
        jobs = [
            Job(max_tokens=self.args.max_tokens, temperature=self.args.temperature, **j)
            for j in jobs_data
        ]
        t0 = time.time()
        await client.publish_jobs(queue, jobs)

        results: List[Result] = []
        done = asyncio.Event()

        async def cb(delivery):
            results.append(Result.model_validate_json(delivery.body))
            await delivery.ack()
            if len(results) >= len(jobs):
                done.set()

        await client.consume_results(queue, cb, prefetch=1000)
        try:
            await asyncio.wait_for(done.wait(), timeout=self.args.timeout)
        except asyncio.TimeoutError:
            print(f"TIMEOUT: {len(results)}/{len(jobs)} results after {self.args.timeout}s")
        total_time = time.time() - t0

        for w in workers:
            if w.poll() is None:
                w.send_signal(signal.SIGTERM)
        await client.disconnect()
        for w in workers:
            try:
                w.wait(timeout=30)
            except subprocess.TimeoutExpired:
                w.kill()
        self.procs = [p for p in self.procs if p.poll() is None]

        in_toks = sum(r.prompt_tokens or 0 for r in results)
        out_toks = sum(r.output_tokens or 0 for r in results)
        lat = [r.duration_ms for r in results]
        waits = [r.queue_wait_ms for r in results if r.queue_wait_ms is not None]
        lat_sorted = sorted(lat) or [0.0]

        def pct(p: float) -> float:
            return lat_sorted[min(len(lat_sorted) - 1, int(p * len(lat_sorted)))]

        return RunResult(
            max_num_seqs=max_num_seqs,
            total_requests=len(jobs),
            successful_requests=len(results),
            failed_requests=len(jobs) - len(results),
            total_time_seconds=total_time,
            total_input_tokens=in_toks,
            total_output_tokens=out_toks,
            input_tokens_per_second=in_toks / total_time,
            output_tokens_per_second=out_toks / total_time,
            total_tokens_per_second=(in_toks + out_toks) / total_time,
            jobs_per_second=len(results) / total_time,
            avg_processing_time_ms=statistics.mean(lat) if lat else 0.0,
            avg_queue_wait_ms=statistics.mean(waits) if waits else 0.0,
            avg_total_latency_ms=statistics.mean(lat) if lat else 0.0,
            p95_total_latency_ms=pct(0.95),
            p99_total_latency_ms=pct(0.99),
        )


def main() -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", default="tower-plus-2b")
    ap.add_argument("--samples", type=int, default=1000)
    ap.add_argument("--batch-sizes", default="128,256,512",
                    help="max_num_seqs sweep (reference default, performance_benchmark.py:645)")
    ap.add_argument("--workers", type=int, default=1)
    ap.add_argument("--gpus", type=int, default=0, help="GPUs to spread workers over (0=CPU)")
    ap.add_argument("--max-tokens", type=int, default=128)
    ap.add_argument("--prompt-tokens", type=int, default=512)
    ap.add_argument("--long-frac", type=float, default=0.0,
                    help="fraction of jobs given a long prompt (mixed workload)")
    ap.add_argument("--long-prompt-tokens", type=int, default=8192)
    ap.add_argument("--temperature", type=float, default=0.7)
    ap.add_argument("--max-model-len", type=int, default=None)
    ap.add_argument("--prefetch", type=int, default=1250,
                    help="queue prefetch (production value, run_dutch_nemotron.slurm:35)")
    ap.add_argument("--engine-overrides", default=None,
                    help="JSON dict of EngineConfig overrides passed to workers")
    ap.add_argument("--timeout", type=float, default=1800.0)
    ap.add_argument("--port", type=int, default=5673)
    ap.add_argument("--log-dir", default="bench_logs")
    ap.add_argument("--output", default="benchmark_results.json")
    args = ap.parse_args()

    os.makedirs(args.log_dir, exist_ok=True)
    batch_sizes = [int(b) for b in args.batch_sizes.split(",")]

    orch = Orchestrator(args)
    results: List[RunResult] = []
    try:
        orch.start_broker()
        for bs in batch_sizes:
            print(f"=== max_num_seqs={bs} ===", flush=True)
            r = asyncio.run(orch.run_point(bs))
            results.append(r)
            print(json.dumps(asdict(r)), flush=True)
    finally:
        orch.stop_all()

    report = {
        "model": args.model,
        "config": {
            "samples": args.samples, "workers": args.workers,
            "max_tokens": args.max_tokens, "prompt_tokens": args.prompt_tokens,
            "prefetch": args.prefetch,
        },
        "gpu_info": gpu_info(),
        "results": [asdict(r) for r in results],
    }
    with open(args.output, "w") as fh:
        json.dump(report, fh, indent=2)
    print(f"wrote {args.output}")
    if results:
        best = max(results, key=lambda r: r.output_tokens_per_second)
        print(f"best: max_num_seqs={best.max_num_seqs} "
              f"{best.output_tokens_per_second:.0f} out-tok/s "
              f"p95={best.p95_total_latency_ms:.0f} ms")


if __name__ == "__main__":
    main()
