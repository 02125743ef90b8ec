"""Wire-format protocol tests (contract: SURVEY.md §2.5)."""

import threading

import pytest

from adversarial_spec_amd.protocol import (
    MAX_RETRIES,
    RETRY_BASE_DELAY,
    CostTracker,
    detect_agreement,
    extract_spec,
    extract_tasks,
    generate_diff,
    get_critique_summary,
)


class TestAgreement:
    def test_detects_marker(self):
        assert detect_agreement("blah [AGREE] blah")

    def test_marker_alone(self):
        assert detect_agreement("[AGREE]")

    def test_no_marker(self):
        assert not detect_agreement("I agree with this")

    def test_case_sensitive(self):
        assert not detect_agreement("[agree]")


class TestExtractSpec:
    def test_basic(self):
        assert extract_spec("x [SPEC]the spec[/SPEC] y") == "the spec"

    def test_strips(self):
        assert extract_spec("[SPEC]\n  body \n[/SPEC]") == "body"

    def test_missing_open(self):
        assert extract_spec("body[/SPEC]") is None

    def test_missing_close(self):
        assert extract_spec("[SPEC]body") is None

    def test_first_pair_wins(self):
        assert extract_spec("[SPEC]a[/SPEC][SPEC]b[/SPEC]") == "a"

    def test_empty(self):
        assert extract_spec("[SPEC][/SPEC]") == ""


class TestExtractTasks:
    def test_single_full_task(self):
        text = """[TASK]
title: Build login
type: user-story
priority: high
description: Implement login flow
acceptance_criteria:
- user can log in
- errors are shown
[/TASK]"""
        tasks = extract_tasks(text)
        assert len(tasks) == 1
        t = tasks[0]
        assert t["title"] == "Build login"
        assert t["type"] == "user-story"
        assert t["priority"] == "high"
        assert t["description"] == "Implement login flow"
        assert t["acceptance_criteria"] == ["user can log in", "errors are shown"]

    def test_multiline_description(self):
        text = "[TASK]\ntitle: T\ndescription: line one\nline two\n[/TASK]"
        tasks = extract_tasks(text)
        assert tasks[0]["description"] == "line one\nline two"

    def test_titleless_dropped(self):
        text = "[TASK]\ndescription: no title\n[/TASK]"
        assert extract_tasks(text) == []

    def test_unclosed_dropped(self):
        assert extract_tasks("[TASK]\ntitle: x\n") == []

    def test_multiple_tasks(self):
        text = "[TASK]\ntitle: a\n[/TASK]\n[TASK]\ntitle: b\n[/TASK]"
        assert [t["title"] for t in extract_tasks(text)] == ["a", "b"]

    def test_criteria_ignores_nonbullet(self):
        text = "[TASK]\ntitle: t\nacceptance_criteria:\n- one\nnot a bullet\n- two\n[/TASK]"
        # non-bullet lines under acceptance_criteria continue... the reference
        # appends them to current_value; our parser treats only "- " lines as
        # items and other lines are appended too per the continuation rule.
        tasks = extract_tasks(text)
        assert "one" in tasks[0]["acceptance_criteria"]
        assert "two" in tasks[0]["acceptance_criteria"]


class TestSummaryAndDiff:
    def test_summary_before_spec(self):
        s = get_critique_summary("critique here [SPEC]spec[/SPEC]")
        assert s == "critique here"

    def test_summary_truncates(self):
        s = get_critique_summary("x" * 500, max_length=100)
        assert len(s) == 103 and s.endswith("...")

    def test_diff_headers(self):
        d = generate_diff("a\nb\n", "a\nc\n")
        assert "--- previous" in d and "+++ current" in d
        assert "-b" in d and "+c" in d

    def test_diff_identical(self):
        assert generate_diff("same\n", "same\n") == ""


class TestConstants:
    def test_max_retries(self):
        assert MAX_RETRIES == 3

    def test_backoff_base(self):
        assert RETRY_BASE_DELAY == 1.0

    def test_backoff_schedule(self):
        delays = [RETRY_BASE_DELAY * (2**k) for k in range(MAX_RETRIES - 1)]
        assert delays == [1.0, 2.0]


class TestCostTracker:
    def test_accumulates(self):
        ct = CostTracker()
        c = ct.add("gpt-4o", 1_000_000, 1_000_000)
        assert c == pytest.approx(12.50)
        assert ct.total_cost == pytest.approx(12.50)
        assert ct.by_model["gpt-4o"]["input_tokens"] == 1_000_000

    def test_default_cost_for_unknown(self):
        ct = CostTracker()
        c = ct.add("unknown-model", 1_000_000, 0)
        assert c == pytest.approx(5.00)

    def test_local_models_free(self):
        ct = CostTracker()
        assert ct.add("local/llama-3-8b", 10_000, 10_000) == 0.0

    def test_thread_safety(self):
        ct = CostTracker()

        def work():
            for _ in range(500):
                ct.add("local/llama-3-8b", 1, 1)

        threads = [threading.Thread(target=work) for _ in range(8)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert ct.total_input_tokens == 4000
        assert ct.by_model["local/llama-3-8b"]["output_tokens"] == 4000

    def test_summary_text(self):
        ct = CostTracker()
        ct.add("gpt-4o", 100, 200)
        s = ct.summary()
        assert "Cost Summary" in s and "gpt-4o" in s
