import json

from opsagent_amd.utils.jsonrepair import clean_json, extract_field, parse_json
from opsagent_amd.utils.perf import PerfStats
from opsagent_amd.utils.yamlextract import extract_yaml


class TestPerfStats:
    def test_timer_and_stats(self):
        p = PerfStats()
        p.start_timer("op")
        p.stop_timer("op")
        s = p.get_metric_stats("op")
        assert s["count"] == 1
        assert s["min"] >= 0

    def test_percentiles(self):
        p = PerfStats()
        for v in range(1, 101):
            p.record_metric("m", float(v))
        s = p.get_metric_stats("m")
        assert s["p50"] == 50.0
        assert s["p95"] == 95.0
        assert s["p99"] == 99.0
        assert s["min"] == 1.0 and s["max"] == 100.0

    def test_trace_and_reset(self):
        p = PerfStats()
        with p.trace("traced"):
            pass
        assert p.get_metric_stats("traced")["count"] == 1
        p.reset()
        assert p.get_metric_stats("traced") is None

    def test_disabled(self):
        p = PerfStats(enabled=False)
        p.record_metric("x", 1.0)
        assert p.get_metric_stats("x") is None

    def test_format_table(self):
        p = PerfStats()
        p.record_metric("aaa", 3.0)
        assert "aaa" in p.format_table()


class TestJsonRepair:
    def test_parse_clean(self):
        assert parse_json('{"a": 1}') == {"a": 1}

    def test_extract_from_prose(self):
        s = 'Here is the result:\n{"final_answer": "done"}\nhope that helps'
        assert parse_json(s) == {"final_answer": "done"}

    def test_code_fence(self):
        s = '```json\n{"a": "b"}\n```'
        assert parse_json(s) == {"a": "b"}

    def test_newline_in_string(self):
        s = '{"a": "line1\nline2"}'
        assert parse_json(s) == {"a": "line1\nline2"}

    def test_trailing_comma(self):
        assert parse_json('{"a": 1,}') == {"a": 1}
        assert parse_json('{"a": [1,2,],}') == {"a": [1, 2]}

    def test_nested_braces_in_string(self):
        s = 'x {"a": "has { brace", "b": 2} y'
        assert parse_json(s) == {"a": "has { brace", "b": 2}

    def test_unrecoverable(self):
        assert parse_json("no json here at all") is None

    def test_extract_field(self):
        assert extract_field('{"final_answer": "42"}', "final_answer") == "42"
        # regex fallback on broken JSON
        broken = '{"thought": "x", "final_answer": "the answer", '
        assert extract_field(broken, "final_answer") == "the answer"

    def test_extract_field_nonstring(self):
        assert json.loads(extract_field('{"action": {"name": "k"}}', "action")) == {"name": "k"}

    def test_clean_json_idempotent(self):
        s = clean_json('{"a": 1}')
        assert clean_json(s) == s


class TestYamlExtract:
    def test_yaml_fence(self):
        text = "here\n```yaml\nkind: Pod\n```\nbye"
        assert extract_yaml(text) == "kind: Pod"

    def test_any_fence(self):
        text = "```\nkind: Service\n```"
        assert extract_yaml(text) == "kind: Service"

    def test_plain(self):
        assert extract_yaml("kind: Deployment") == "kind: Deployment"


def test_perf_auto_reset():
    import time

    from opsagent_amd.utils.perf import start_auto_reset

    p = PerfStats()
    p.record_metric("x", 1.0)
    stop = start_auto_reset(p, 0.05)
    try:
        time.sleep(0.2)
        assert p.get_metric_stats("x") is None
    finally:
        stop.set()


def test_log_file_rotation(tmp_path):
    """File sink writes into the daily-named log file with rotation
    configured (ref utils/logger.go lumberjack + daily naming)."""
    import importlib
    import logging as _logging

    from opsagent_amd.utils import logging as olog

    olog.init_logging(level="info", fmt="json", output="file", log_dir=str(tmp_path))
    log = olog.get_logger("rotation-test")
    log.info("hello rotation")
    for h in _logging.getLogger().handlers + _logging.getLogger("opsagent").handlers:
        try:
            h.flush()
        except Exception:
            pass
    files = list(tmp_path.glob("*.log"))
    assert files, "no log file created"
    content = "".join(f.read_text() for f in files)
    assert "hello rotation" in content
    # restore stderr logging for the rest of the suite
    olog.init_logging(level="warning", fmt="console", output="stderr")


def test_jsonrepair_mutation_corpus():
    """CleanJSON-style repair survives the reference's failure modes: fenced
    blocks, trailing commas, raw newlines in strings, single quotes, and
    leading/trailing prose (ref pkg/utils/json.go:16-120)."""
    import json

    from opsagent_amd.utils.jsonrepair import parse_json

    cases = [
        'Sure! Here is the JSON:\n```json\n{"a": 1}\n```\nHope that helps.',
        '{"a": 1, "b": [1, 2,], }',
        '{"a": "line one\nline two"}',
        "{'a': 'single'}",
        'prefix {"a": {"b": 2}} suffix',
        '{"a": 1} {"ignored": 2}',
    ]
    for c in cases:
        obj = parse_json(c)
        assert isinstance(obj, dict) and "a" in obj, c

    # irreparable input returns None rather than raising
    assert parse_json("no json here at all") is None


def test_count_tokens_message_overhead():
    """Per-message overhead is counted (ref tokens.go:60-107): two messages
    cost more than the sum of their bare contents."""
    from opsagent_amd.llm.tokens import count_tokens

    one = count_tokens([{"role": "user", "content": "hello"}])
    two = count_tokens(
        [{"role": "user", "content": "hello"}, {"role": "assistant", "content": ""}]
    )
    assert two > one > 0


def test_perf_trace_exception_safe():
    """perf.trace records the timer even when the body raises (defer
    semantics of the ref's TraceFunc)."""
    import pytest

    p = PerfStats()
    p.enabled = True
    with pytest.raises(ValueError):
        with p.trace("boom_op"):
            raise ValueError("x")
    stats = p.get_stats()
    assert "boom_op" in stats and stats["boom_op"]["count"] == 1
