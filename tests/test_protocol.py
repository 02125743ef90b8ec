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


class TestMutationKillers:
    """Kill-the-mutant tests for survivors of tools/mutation_check.py
    (mirrors the reference's mutmut-driven test additions,
    test_models.py:856-872 class of tests)."""

    def test_model_response_agreed_defaults_false(self):
        from adversarial_spec_amd.protocol import ModelResponse

        assert ModelResponse(model="m").agreed is False

    def test_summary_truncates_at_exactly_300_default(self):
        # 300 chars pass through untouched; 301 gains the ellipsis
        assert get_critique_summary("x" * 300) == "x" * 300
        assert get_critique_summary("x" * 301) == "x" * 300 + "..."

    def test_summary_spec_tag_at_position_zero_keeps_response(self):
        # spec_start == 0 must NOT slice to the empty prefix
        r = "[SPEC]\nbody\n[/SPEC]"
        assert get_critique_summary(r) == r

    def test_summary_spec_tag_mid_response(self):
        r = "critique text\n[SPEC]\nbody\n[/SPEC]"
        assert get_critique_summary(r) == "critique text"

    def test_task_single_line_value_is_verbatim(self):
        # len(buf) == 1 path: value taken as-is (no join/strip rewrite)
        t = extract_tasks(
            "[TASK]\ntitle: Exact Title\ntype: feature\n"
            "description: one liner\nacceptance_criteria:\n- a\n[/TASK]"
        )
        assert t[0]["title"] == "Exact Title"
        assert t[0]["description"] == "one liner"

    def test_task_two_line_value_joins_both(self):
        # len(buf) == 2 crosses the >1 boundary: both lines must survive
        t = extract_tasks(
            "[TASK]\ntitle: T\ndescription: first\nsecond\n"
            "type: chore\n[/TASK]"
        )
        assert t[0]["description"] == "first\nsecond"

    def test_task_field_value_offset_exact(self):
        # rest = line[len(key)+1:]: off-by-one would eat or keep the colon
        t = extract_tasks("[TASK]\ntitle:NoSpace\ntype: bug\n[/TASK]")
        assert t[0]["title"] == "NoSpace"
        t2 = extract_tasks("[TASK]\ntitle:  padded\ntype: bug\n[/TASK]")
        assert t2[0]["title"] == "padded"
