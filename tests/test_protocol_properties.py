"""Property-based tests (hypothesis) for the wire-format surface.

The reference pins these contracts with example-based tests
(models.py:149-247 parsers); properties cover the input space the
examples cannot: arbitrary text around markers, adversarial whitespace,
marker fragments, and chunking invariants.
"""

from __future__ import annotations

from hypothesis import given, settings
from hypothesis import strategies as st

from adversarial_spec_amd.protocol import (
    detect_agreement,
    extract_spec,
    extract_tasks,
    get_critique_summary,
)
from adversarial_spec_amd.telegram import split_message

# text WITHOUT protocol markers (so properties can add them precisely)
clean_text = st.text(
    alphabet=st.characters(blacklist_characters="[]"), max_size=400
)


class TestAgreementProperties:
    @given(pre=clean_text, post=clean_text)
    @settings(max_examples=60, deadline=None)
    def test_marker_always_detected(self, pre, post):
        assert detect_agreement(pre + "[AGREE]" + post)

    @given(text=clean_text)
    @settings(max_examples=60, deadline=None)
    def test_no_marker_never_detected(self, text):
        assert not detect_agreement(text)


class TestSpecProperties:
    @given(pre=clean_text, body=clean_text, post=clean_text)
    @settings(max_examples=60, deadline=None)
    def test_round_trip_strips_body(self, pre, body, post):
        got = extract_spec(pre + "[SPEC]" + body + "[/SPEC]" + post)
        assert got == body.strip()

    @given(text=clean_text)
    @settings(max_examples=60, deadline=None)
    def test_missing_tags_none(self, text):
        assert extract_spec(text) is None
        assert extract_spec("[SPEC]" + text) is None
        assert extract_spec(text + "[/SPEC]") is None

    @given(body=clean_text)
    @settings(max_examples=60, deadline=None)
    def test_summary_never_exceeds_budget(self, body):
        s = get_critique_summary(body + "[SPEC]x[/SPEC]", max_length=50)
        assert len(s) <= 53  # 50 + "..."


class TestTaskProperties:
    titles = st.text(
        alphabet=st.characters(
            whitelist_categories=("Lu", "Ll", "Nd"), max_codepoint=0x7F
        ),
        min_size=1,
        max_size=40,
    )

    @given(title=titles, desc=titles)
    @settings(max_examples=60, deadline=None)
    def test_minimal_task_parses(self, title, desc):
        text = f"[TASK]\ntitle: {title}\ndescription: {desc}\n[/TASK]"
        tasks = extract_tasks(text)
        assert len(tasks) == 1
        assert tasks[0]["title"] == title.strip()
        assert tasks[0]["description"] == desc.strip()

    @given(titles_list=st.lists(titles, min_size=1, max_size=5))
    @settings(max_examples=40, deadline=None)
    def test_n_blocks_n_tasks(self, titles_list):
        text = "\n".join(
            f"[TASK]\ntitle: {t}\n[/TASK]" for t in titles_list
        )
        assert len(extract_tasks(text)) == len(titles_list)

    @given(noise=clean_text)
    @settings(max_examples=60, deadline=None)
    def test_untitled_blocks_dropped_and_never_crash(self, noise):
        # arbitrary block bodies parse without raising; no title -> dropped
        tasks = extract_tasks("[TASK]\n" + noise + "\n[/TASK]")
        for t in tasks:
            assert t.get("title")

    @given(items=st.lists(titles, min_size=1, max_size=6))
    @settings(max_examples=40, deadline=None)
    def test_acceptance_criteria_items_preserved(self, items):
        lines = "\n".join(f"- {i}" for i in items)
        text = f"[TASK]\ntitle: t\nacceptance_criteria:\n{lines}\n[/TASK]"
        tasks = extract_tasks(text)
        assert tasks[0]["acceptance_criteria"] == [i.strip() for i in items]


class TestSplitMessageProperties:
    @given(text=st.text(max_size=3000), limit=st.integers(10, 200))
    @settings(max_examples=80, deadline=None)
    def test_chunks_respect_limit(self, text, limit):
        for c in split_message(text, limit):
            assert len(c) <= limit

    @given(text=st.text(alphabet=st.characters(
        blacklist_characters="\n"), min_size=1, max_size=2000),
        limit=st.integers(10, 100))
    @settings(max_examples=80, deadline=None)
    def test_no_newline_content_preserved_exactly(self, text, limit):
        # without newline cuts, concatenation is the identity
        assert "".join(split_message(text, limit)) == text

    @given(text=st.text(min_size=1, max_size=2000), limit=st.integers(10, 100))
    @settings(max_examples=80, deadline=None)
    def test_content_preserved_up_to_cut_newlines(self, text, limit):
        # newline-boundary cuts drop ONLY the boundary newlines
        joined = "".join(split_message(text, limit))
        assert joined.replace("\n", "") == text.replace("\n", "")

    @given(text=st.text(min_size=1, max_size=2000), limit=st.integers(10, 100))
    @settings(max_examples=80, deadline=None)
    def test_no_empty_chunks(self, text, limit):
        # an empty chunk would become an empty Telegram send
        for c in split_message(text, limit):
            assert c != ""
