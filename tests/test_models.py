"""Wire-model tests (coverage modeled on reference tests/test_models.py)."""

import json

import pytest
from pydantic import ValidationError

from llmq_amd.core.models import ErrorInfo, Job, QueueStats, Result, WorkerHealth, utcnow


class TestJob:
    def test_prompt_job(self):
        job = Job(id="a", prompt="hello {name}", name="world")
        assert job.get_formatted_prompt() == "hello world"

    def test_messages_job(self):
        job = Job(id="a", messages=[{"role": "user", "content": "hi"}])
        assert job.messages[0]["content"] == "hi"

    def test_both_prompt_and_messages_rejected(self):
        with pytest.raises(ValidationError):
            Job(id="a", prompt="x", messages=[{"role": "user", "content": "y"}])

    def test_neither_rejected(self):
        with pytest.raises(ValidationError):
            Job(id="a")

    def test_extra_fields_passthrough(self):
        job = Job(id="a", prompt="p", source_lang="nl", score=0.5)
        extra = job.extra_fields()
        assert extra == {"source_lang": "nl", "score": 0.5}

    def test_braces_in_data_survive(self):
        job = Job(id="a", prompt="eval {code}", code="f({x: 1})")
        assert job.get_formatted_prompt() == "eval f({x: 1})"

    def test_template_without_field_left_verbatim(self):
        job = Job(id="a", prompt="no placeholder here")
        assert job.get_formatted_prompt() == "no placeholder here"

    def test_stop_sequences(self):
        job = Job(id="a", prompt="p", stop=["\n\n", "END"])
        assert job.stop == ["\n\n", "END"]

    def test_sampling_overrides(self):
        job = Job(id="a", prompt="p", temperature=0.1, top_p=0.9, max_tokens=32)
        assert job.temperature == 0.1
        assert job.top_p == 0.9
        assert job.max_tokens == 32

    def test_json_roundtrip(self):
        job = Job(id="a", prompt="hello {x}", x="y", chat_mode=False)
        blob = job.model_dump_json()
        back = Job.model_validate_json(blob)
        assert back.id == job.id
        assert back.extra_fields() == {"x": "y"}

    def test_wire_format_matches_reference(self):
        # The on-wire JSON keys the reference emits must be accepted.
        wire = json.dumps(
            {"id": "j1", "prompt": "p {a}", "chat_mode": False, "stop": None, "a": "b"}
        )
        job = Job.model_validate_json(wire)
        assert job.get_formatted_prompt() == "p b"


class TestResult:
    def test_basic(self):
        r = Result(id="a", prompt="p", result="out", worker_id="w", duration_ms=12.5)
        assert r.timestamp is not None

    def test_extra_passthrough(self):
        r = Result(
            id="a", prompt="p", result="out", worker_id="w", duration_ms=1.0, lang="de"
        )
        assert json.loads(r.model_dump_json())["lang"] == "de"

    def test_engine_timing_fields(self):
        r = Result(
            id="a", prompt="p", result="o", worker_id="w", duration_ms=1.0,
            prefill_ms=5.0, decode_ms=20.0, prompt_tokens=10, output_tokens=50,
            finish_reason="eos",
        )
        assert r.output_tokens == 50


class TestStatsModels:
    def test_queue_stats_defaults(self):
        s = QueueStats(queue_name="q")
        assert s.stats_source == "broker"
        assert s.message_count is None

    def test_worker_health(self):
        h = WorkerHealth(
            worker_id="w", status="active", last_seen=utcnow(), jobs_processed=3
        )
        assert h.jobs_processed == 3

    def test_error_info(self):
        e = ErrorInfo(job_id="a", error_message="boom", timestamp=utcnow())
        assert e.worker_id is None


def test_reference_env_names_accepted(monkeypatch):
    """A .env written for the reference (RABBITMQ_URL / VLLM_*) configures
    this framework unchanged (reference config.py:10-44); LLMQ_* wins when
    both are set."""
    from llmq_amd.core.config import Config

    monkeypatch.setenv("RABBITMQ_URL", "amqp://u:p@h:5672/vh")
    monkeypatch.setenv("VLLM_QUEUE_PREFETCH", "1250")
    monkeypatch.setenv("VLLM_MAX_TOKENS", "4096")
    cfg = Config()
    assert cfg.broker_url == "amqp://u:p@h:5672/vh"
    assert cfg.queue_prefetch == 1250
    assert cfg.max_tokens == 4096
    monkeypatch.setenv("LLMQ_QUEUE_PREFETCH", "64")
    assert Config().queue_prefetch == 64


def test_map_json_and_template_modes():
    """--map value forms from the reference (submit.py:184-236): JSON
    templates build chat messages from rows; {var} strings interpolate; a
    bare `text` column falls back to prompt."""
    from llmq_amd.utils.template import create_job_from_data

    row = {"text": "dirty <html>", "lang": "nl"}
    j = create_job_from_data(
        row, "d1", None,
        {"messages": '[{"role": "user", "content": "Clean: {text} ({lang})"}]'})
    assert j.messages == [{"role": "user", "content": "Clean: dirty <html> (nl)"}]
    assert j.prompt is None

    j2 = create_job_from_data(row, "d2", None, {"prompt": "Translate {text} to {lang}"})
    assert j2.prompt == "Translate dirty <html> to nl"

    j3 = create_job_from_data({"text": "as-is"}, "d3", None, None)
    assert j3.prompt == "as-is"

    import pytest as _pytest
    with _pytest.raises(ValueError, match="invalid JSON"):
        create_job_from_data(row, "d4", None, {"messages": '[{"role": broken}]'})
