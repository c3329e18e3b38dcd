"""Tolerant section-parser contracts (the REGEXP_EXTRACT columns of the
lab CTAS statements — LAB1-Walkthrough.md:202-204, LAB3:462-464,
LAB4:405-408): markdown-bold-optional headers, missing sections yield '',
Flink REGEXP_EXTRACT returns '' on no match."""

import json

from quickstart_streaming_agents_amd.agents.parse import (
    LAB4_VERDICTS, parse_lab1_sections, parse_lab3_sections,
    parse_lab4_sections, regexp_extract)


def test_regexp_extract_flink_semantics():
    assert regexp_extract("price: $12.50", r"\$(\d+\.\d+)") == "12.50"
    assert regexp_extract("no digits", r"(\d+)") == ""          # no match
    assert regexp_extract(None, r"(x)") == ""                   # null subject
    assert regexp_extract("ab", r"a(x)?b") == ""                # None group


def test_lab1_sections_plain_and_bold():
    plain = ("Competitor Price: $209.99\n"
             "Decision: MATCH\n"
             "Summary: Competitor undercuts; matched and emailed.")
    bold = plain.replace("Competitor Price:", "**Competitor Price:**") \
                .replace("Decision:", "**Decision:**") \
                .replace("Summary:", "**Summary:**")
    for resp in (plain, bold):
        s = parse_lab1_sections(resp)
        assert s["competitor_price"] == "$209.99"
        assert s["decision"] == "MATCH"
        assert s["summary"].startswith("Competitor undercuts")


def test_lab1_missing_sections_empty():
    s = parse_lab1_sections("model rambled with no format at all")
    assert s == {"competitor_price": "", "decision": "", "summary": ""}


def test_lab3_sections_with_json_fence():
    resp = ("Dispatch Summary:\nSending 3 boats to French Quarter.\n\n"
            "Dispatch JSON:\n```json\n"
            '{"boats": ["BOAT-01", "BOAT-03", "BOAT-07"]}\n```\n\n'
            "API Response:\n```json\n"
            '{"status": "dispatched", "count": 3}\n```')
    s = parse_lab3_sections(resp)
    assert "French Quarter" in s["dispatch_summary"]
    assert json.loads(s["dispatch_json"])["boats"] == [
        "BOAT-01", "BOAT-03", "BOAT-07"]
    assert json.loads(s["api_response"])["count"] == 3


def test_lab4_sections_and_verdict_enum():
    resp = ("Verdict: DENY_FRAUD\n"
            "Issues Found:\n- duplicate address\n- inflated amount\n"
            "Policy Basis:\nFEMA IA policy 4.2\n"
            "Summary:\nClaim denied for fraud indicators.")
    s = parse_lab4_sections(resp)
    assert s["verdict"] in LAB4_VERDICTS
    assert s["verdict"] == "DENY_FRAUD"
    assert "duplicate address" in s["issues_found"]
    assert s["policy_basis"] == "FEMA IA policy 4.2"
    assert s["summary"].startswith("Claim denied")
