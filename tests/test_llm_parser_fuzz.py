"""Property fuzz of the LLM output parsers: on ARBITRARY text every
parser either returns a well-formed structure or raises ParseError —
never any other exception (the orchestrator's fallback ladder catches
exactly ParseError; anything else would crash a phase). Mirrors the
reference's malformed-JSON salvage tests (llm-parser.test.ts) with
generated adversarial input instead of a fixed list.
"""
from __future__ import annotations

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from runbookai_amd.agent.llm_parser import (  # noqa: E402
    ParseError,
    extract_json,
    parse_conclusion,
    parse_evidence_evaluation,
    parse_hypothesis_generation,
    parse_json,
    parse_remediation_plan,
    parse_triage_response,
)

PARSERS = [parse_json, parse_triage_response, parse_hypothesis_generation,
           parse_evidence_evaluation, parse_conclusion, parse_remediation_plan]

# adversarial soup: JSON shards, quotes, braces, unicode, control chars
soup = st.text(
    alphabet=st.sampled_from(list('{}[]",:truefalsenull0123456789.eE+- \n\t')
                             + ["ä", "漢", "\\", "'", "x"]),
    min_size=0, max_size=120)
jsonish = st.recursive(
    st.one_of(st.none(), st.booleans(), st.integers(-1e6, 1e6),
              st.text(max_size=12)),
    lambda inner: st.one_of(st.lists(inner, max_size=4),
                            st.dictionaries(st.text(max_size=8), inner, max_size=4)),
    max_leaves=8)


@settings(max_examples=150, deadline=None)
@given(text=soup)
def test_parsers_raise_only_parse_error_on_soup(text):
    for p in PARSERS:
        try:
            out = p(text)
        except ParseError:
            continue
        assert out is not None or p is parse_json


@settings(max_examples=100, deadline=None)
@given(value=jsonish, prefix=soup, suffix=soup)
def test_embedded_json_extraction(value, prefix, suffix):
    """JSON embedded in arbitrary prose: extract_json finds a parseable
    region whenever the payload is an object/array not confused by the
    surrounding soup's own brackets."""
    import json

    payload = json.dumps(value)
    if not isinstance(value, (dict, list)):
        return
    clean_prefix = prefix.replace("{", "").replace("[", "")
    text = clean_prefix + payload + suffix
    try:
        got = parse_json(text)
    except ParseError:
        return   # allowed: suffix soup can break bracket matching
    assert isinstance(got, (dict, list))


@settings(max_examples=100, deadline=None)
@given(d=st.dictionaries(st.text(max_size=8), st.text(max_size=8), max_size=5))
def test_clean_objects_always_parse(d):
    import json

    got = parse_json(json.dumps(d))
    assert got == d


@settings(max_examples=60, deadline=None)
@given(text=soup)
def test_extract_json_output_is_substring_or_error(text):
    try:
        frag = extract_json(text)
    except ParseError:
        return
    assert frag in text or frag.strip() in text
