"""llm-parser schema round-trips incl. malformed JSON salvage
(parity with reference agent/__tests__/llm-parser.test.ts, 561 LoC)."""
import pytest

from runbookai_amd.agent.llm_parser import (
    ParseError,
    extract_json,
    fill_prompt,
    parse_conclusion,
    parse_evidence_evaluation,
    parse_hypothesis_generation,
    parse_log_analysis,
    parse_remediation_plan,
    parse_triage_response,
    split_schema_tag,
)


class TestExtractJSON:
    def test_fenced(self):
        assert extract_json('Here:\n```json\n{"a": 1}\n```\ndone') == '{"a": 1}'

    def test_fence_without_lang(self):
        assert extract_json('```\n{"a": 1}\n```') == '{"a": 1}'

    def test_braces_in_prose(self):
        assert extract_json('The answer is {"a": {"b": 2}} ok?') == '{"a": {"b": 2}}'

    def test_array(self):
        assert extract_json('list: [1, 2, 3] end') == "[1, 2, 3]"

    def test_braces_inside_strings(self):
        s = '{"a": "has } brace"}'
        assert extract_json("x " + s + " y") == s

    def test_empty_raises(self):
        with pytest.raises(ParseError):
            extract_json("")


class TestTriage:
    def test_valid(self):
        t = parse_triage_response(
            '{"summary": "s", "symptoms": ["latency"], "affectedServices": ["api"],'
            ' "severity": "high", "timeline": "09:10"}'
        )
        assert t["severity"] == "high"
        assert t["symptoms"] == ["latency"]

    def test_bad_severity_defaults(self):
        t = parse_triage_response('{"summary": "s", "severity": "EXTREME"}')
        assert t["severity"] == "medium"

    def test_trailing_comma_salvage(self):
        t = parse_triage_response('{"summary": "s", "symptoms": ["a",],}')
        assert t["summary"] == "s"


class TestHypothesisGeneration:
    def test_caps_at_five(self):
        items = [{"statement": f"h{i}", "rationale": "", "priority": 1} for i in range(8)]
        import json

        out = parse_hypothesis_generation(json.dumps({"hypotheses": items}))
        assert len(out) == 5

    def test_bare_array_accepted(self):
        out = parse_hypothesis_generation('[{"statement": "x", "rationale": "y", "priority": 9}]')
        assert out[0]["priority"] == 5  # clamped

    def test_invalid_items_skipped(self):
        out = parse_hypothesis_generation(
            '{"hypotheses": [{"nope": 1}, {"statement": "good", "rationale": "r", "priority": 2}]}'
        )
        assert len(out) == 1

    def test_all_invalid_raises(self):
        with pytest.raises(ParseError):
            parse_hypothesis_generation('{"hypotheses": [{"x": 1}]}')


class TestEvidenceEvaluation:
    def test_valid_branch(self):
        e = parse_evidence_evaluation(
            '{"action": "branch", "confidence": 0.6, "reasoning": "split",'
            ' "evidence": [{"description": "d", "supports": true}],'
            ' "subHypotheses": [{"statement": "s", "rationale": "r", "priority": 1}]}'
        )
        assert e["action"] == "branch"
        assert len(e["subHypotheses"]) == 1

    def test_unknown_action_becomes_continue(self):
        e = parse_evidence_evaluation('{"action": "explode", "confidence": 0.5, "reasoning": ""}')
        assert e["action"] == "continue"

    def test_string_evidence_normalized(self):
        e = parse_evidence_evaluation(
            '{"action": "confirm", "confidence": 2.5, "reasoning": "", "evidence": ["saw errors"]}'
        )
        assert e["confidence"] == 1.0
        assert e["evidence"][0]["description"] == "saw errors"


class TestConclusion:
    def test_valid(self):
        c = parse_conclusion(
            '{"rootCause": "redis pool exhausted", "confidence": "high", "summary": "s"}'
        )
        assert c["rootCause"] == "redis pool exhausted"

    def test_missing_root_cause_raises(self):
        with pytest.raises(ParseError):
            parse_conclusion('{"confidence": "high", "summary": "s"}')

    def test_snake_case_accepted(self):
        c = parse_conclusion('{"root_cause": "x", "confidence": "nope", "summary": ""}')
        assert c["rootCause"] == "x"
        assert c["confidence"] == "medium"


class TestRemediation:
    def test_steps_normalized(self):
        p = parse_remediation_plan(
            '{"summary": "fix", "steps": ["restart service",'
            ' {"description": "delete node", "risk": "critical"}], "rollback": "undo"}'
        )
        assert p["steps"][0]["description"] == "restart service"
        assert p["steps"][0]["risk"] == "low"
        assert p["steps"][1]["requiresApproval"] is True

    def test_requires_approval_default_by_risk(self):
        p = parse_remediation_plan('{"summary": "", "steps": [{"description": "d", "risk": "high"}]}')
        assert p["steps"][0]["requiresApproval"] is True


class TestLogAnalysis:
    def test_valid(self):
        r = parse_log_analysis(
            '{"summary": "s", "patterns": [{"pattern": "oom", "count": 3, "severity": "critical"}],'
            ' "services": ["api"]}'
        )
        assert r["patterns"][0]["count"] == 3


class TestPromptFill:
    def test_fill_and_schema_tag(self):
        p = fill_prompt("triage", query="q", context="ctx")
        kind, body = split_schema_tag(p)
        assert kind == "triage"
        assert "Incident query: q" in body
        assert '"summary"' in body

    def test_missing_fields_tolerated(self):
        p = fill_prompt("generateConclusion", tag_schema=False, summary="s")
        assert "Confirmed hypotheses" in p

    def test_untagged(self):
        p = fill_prompt("triage", tag_schema=False, query="q", context="c")
        kind, body = split_schema_tag(p)
        assert kind is None
        assert body == p


class TestNormalizationEdges:
    """Reference llm-parser.test.ts:104-156, 178-230, 253-320, 522-560."""

    def test_priority_clamped_into_range(self):
        out = parse_hypothesis_generation(
            '{"hypotheses": [{"statement": "a", "rationale": "r", "priority": 99},'
            '{"statement": "b", "rationale": "r", "priority": -3},'
            '{"statement": "c", "rationale": "r", "priority": "bogus"}]}')
        assert [h["priority"] for h in out] == [5, 1, 3]

    def test_confidence_clamped(self):
        from runbookai_amd.agent.llm_parser import parse_evidence_evaluation

        hi = parse_evidence_evaluation('{"action": "confirm", "confidence": 4.2, "reasoning": ""}')
        lo = parse_evidence_evaluation('{"action": "prune", "confidence": -1, "reasoning": ""}')
        assert hi["confidence"] == 1.0 and lo["confidence"] == 0.0

    def test_evaluation_with_sub_hypotheses(self):
        from runbookai_amd.agent.llm_parser import parse_evidence_evaluation

        out = parse_evidence_evaluation(
            '{"action": "branch", "confidence": 0.5, "reasoning": "split",'
            '"subHypotheses": [{"statement": "s1", "rationale": "r", "priority": 2},'
            '{"rationale": "no statement"}]}')
        assert len(out["subHypotheses"]) == 1
        assert out["subHypotheses"][0]["statement"] == "s1"

    def test_triage_initial_hypotheses(self):
        from runbookai_amd.agent.llm_parser import parse_triage_response

        out = parse_triage_response(
            '{"summary": "s", "severity": "high",'
            '"initialHypotheses": [{"statement": "bad deploy", "rationale": "recent"}]}')
        assert out["initialHypotheses"][0]["statement"] == "bad deploy"

    def test_conclusion_string_fields_coerced_to_lists(self):
        from runbookai_amd.agent.llm_parser import parse_conclusion

        out = parse_conclusion(
            '{"rootCause": "rc", "confidence": "high", "summary": "s",'
            '"evidence": "a single string", "contributingFactors": "also one"}')
        assert out["evidence"] == ["a single string"]
        assert out["contributingFactors"] == ["also one"]

    def test_nested_json_extracted(self):
        from runbookai_amd.agent.llm_parser import parse_json

        out = parse_json('Before text {"a": {"b": {"c": [1, 2, {"d": 3}]}}} after')
        assert out["a"]["b"]["c"][2]["d"] == 3

    def test_fill_multiple_occurrences(self):
        from runbookai_amd.agent.llm_parser import PROMPTS, fill_prompt

        PROMPTS["_twice"] = "service {svc} then {svc} again"
        try:
            out = fill_prompt("_twice", tag_schema=False, svc="api")
            assert out.count("api") == 2
        finally:
            del PROMPTS["_twice"]

    def test_all_templates_have_placeholders(self):
        import re

        from runbookai_amd.agent.llm_parser import PROMPTS

        for kind in ("triage", "generateHypotheses", "evaluateEvidence",
                     "generateConclusion", "generateRemediation", "analyzeLogs"):
            assert kind in PROMPTS
            # every template has at least one single-brace placeholder
            assert re.search(r"(?<!\{)\{[a-zA-Z_]+\}(?!\})", PROMPTS[kind]), kind
