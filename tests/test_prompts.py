"""Prompt library contract tests (keys/placeholders/protocol markers)."""

from adversarial_spec_amd.prompts import (
    EXPORT_TASKS_PROMPT,
    FOCUS_AREAS,
    PERSONAS,
    PRESERVE_INTENT_PROMPT,
    PRESS_PROMPT_TEMPLATE,
    REVIEW_PROMPT_TEMPLATE,
    SYSTEM_PROMPT_GENERIC,
    SYSTEM_PROMPT_PRD,
    SYSTEM_PROMPT_TECH,
    get_doc_type_name,
    get_system_prompt,
)


def test_focus_area_keys():
    assert set(FOCUS_AREAS) == {
        "security", "scalability", "performance", "ux", "reliability", "cost"
    }


def test_persona_keys():
    assert set(PERSONAS) == {
        "security-engineer", "oncall-engineer", "junior-developer", "qa-engineer",
        "site-reliability", "product-manager", "data-engineer", "mobile-developer",
        "accessibility-specialist", "legal-compliance",
    }


def test_system_prompts_carry_protocol():
    for p in (SYSTEM_PROMPT_PRD, SYSTEM_PROMPT_TECH, SYSTEM_PROMPT_GENERIC):
        assert "[AGREE]" in p
        assert "[SPEC]" in p and "[/SPEC]" in p


def test_get_system_prompt_by_doc_type():
    assert get_system_prompt("prd") == SYSTEM_PROMPT_PRD
    assert get_system_prompt("tech") == SYSTEM_PROMPT_TECH
    assert get_system_prompt("other") == SYSTEM_PROMPT_GENERIC


def test_persona_normalization():
    direct = get_system_prompt("tech", "security-engineer")
    spaced = get_system_prompt("tech", "Security Engineer")
    underscored = get_system_prompt("tech", "security_engineer")
    assert direct == spaced == underscored == PERSONAS["security-engineer"]


def test_unknown_persona_wrapped():
    p = get_system_prompt("tech", "marine biologist")
    assert "marine biologist" in p


def test_review_template_placeholders():
    msg = REVIEW_PROMPT_TEMPLATE.format(
        round=3, doc_type_name="Technical Specification", spec="SPEC BODY",
        focus_section="FOCUS", context_section="CTX",
    )
    assert "round 3" in msg and "SPEC BODY" in msg and "FOCUS" in msg and "CTX" in msg
    assert "[AGREE]" in msg


def test_press_template_placeholders():
    msg = PRESS_PROMPT_TEMPLATE.format(
        round=2, doc_type_name="PRD x", spec="BODY", context_section="",
    )
    assert "round 2" in msg and "BODY" in msg
    assert "[AGREE]" in msg and "[SPEC]" in msg


def test_export_tasks_grammar_fields():
    msg = EXPORT_TASKS_PROMPT.format(doc_type_name="spec", spec="BODY")
    for field in ("title:", "type:", "priority:", "description:", "acceptance_criteria:"):
        assert field in msg
    assert "[TASK]" in msg and "[/TASK]" in msg


def test_preserve_intent_taxonomy():
    for bucket in ("ERRORS", "RISKS", "PREFERENCES"):
        assert bucket in PRESERVE_INTENT_PROMPT


def test_doc_type_names():
    assert get_doc_type_name("prd") == "Product Requirements Document"
    assert get_doc_type_name("tech") == "Technical Specification"
    assert get_doc_type_name("x") == "specification"
