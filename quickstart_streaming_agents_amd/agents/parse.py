"""REGEXP_EXTRACT + the tolerant section parsers the lab pipelines use.

Implements Flink's REGEXP_EXTRACT(subject, pattern, group) semantics
(returns '' on no match — downstream columns are STRING) and the per-lab
section extraction patterns (LAB1-Walkthrough.md:202-204,
LAB3-Walkthrough.md:462-464, LAB4-Walkthrough.md:405-408): markdown-bold
tolerant ``\\*{0,2}Section:\\*{0,2}`` headers over free-form LLM output.
"""

from __future__ import annotations

import re

LAB4_VERDICTS = ("APPROVE", "APPROVE_PARTIAL", "REQUEST_DOCS",
                 "DENY_INELIGIBLE", "DENY_FRAUD")


def regexp_extract(subject: str, pattern: str, group: int = 1) -> str:
    """Flink REGEXP_EXTRACT: first match's group, '' when no match."""
    m = re.search(pattern, subject or "")
    if not m:
        return ""
    g = m.group(group)
    return g if g is not None else ""


def _sec(subject: str, pattern: str) -> str:
    return regexp_extract(subject, pattern, 1).strip()


def parse_lab1_sections(response: str) -> dict:
    """Competitor Price / Decision / Summary (LAB1-Walkthrough.md:202-204)."""
    return {
        "competitor_price": _sec(
            response,
            r"\*{0,2}Competitor Price:\*{0,2}\s*\n?([\s\S]+?)(?=\n\*{0,2}(?:Decision|Summary):|$)"),
        "decision": _sec(response, r"\*{0,2}Decision:\*{0,2}\s*\n?([A-Z_]+)"),
        "summary": _sec(response, r"\*{0,2}Summary:\*{0,2}\s*\n?([\s\S]+?)$"),
    }


def parse_lab3_sections(response: str) -> dict:
    """Dispatch Summary / Dispatch JSON / API Response (LAB3:462-464)."""
    return {
        "dispatch_summary": _sec(
            response,
            r"\*{0,2}Dispatch Summary:\*{0,2}\s*\n([\s\S]+?)(?=\n\n\*{0,2}Dispatch JSON:\*{0,2})"),
        "dispatch_json": _sec(
            response,
            r"\*{0,2}Dispatch JSON:\*{0,2}\s*\n(?:```json\s*)?([\s\S]+?)(?:```)?(?=\n\n\*{0,2}API Response:\*{0,2})"),
        "api_response": _sec(
            response,
            r"\*{0,2}API Response:\*{0,2}\s*\n(?:```json\s*)?([\s\S]+?)(?:```)?$"),
    }


def parse_lab4_sections(response: str) -> dict:
    """Verdict / Issues Found / Policy Basis / Summary (LAB4:405-408)."""
    return {
        "verdict": _sec(response, r"\*{0,2}Verdict:\*{0,2}\s*([A-Z_]+)"),
        "issues_found": _sec(
            response,
            r"\*{0,2}Issues Found:\*{0,2}\n([\s\S]+?)(?=\n\*{0,2}(?:Policy Basis|Summary|Verdict):|$)"),
        "policy_basis": _sec(
            response,
            r"\*{0,2}Policy Basis:\*{0,2}\n([\s\S]+?)(?=\n\*{0,2}(?:Summary|Verdict):|$)"),
        "summary": _sec(response, r"\*{0,2}Summary:\*{0,2}\n([\s\S]+?)$"),
    }
