"""JSON grammar FSM tests: every PROMPT_SCHEMAS schema must yield valid,
schema-conformant JSON under arbitrary (adversarial/random) byte choices."""
import json
import random

import pytest

from runbookai_amd.agent.llm_parser import (
    PROMPT_SCHEMAS,
    parse_conclusion,
    parse_evidence_evaluation,
    parse_hypothesis_generation,
    parse_triage_response,
)
from runbookai_amd.engine.json_fsm import NUMBER_CLOSE_SENTINEL, JsonFsm, generate_minimal


def drive(schema, chooser, max_bytes=8192):
    return generate_minimal(schema, chooser, max_bytes)


def random_chooser(seed):
    rng = random.Random(seed)
    return lambda allowed: rng.choice(allowed)


class TestBasics:
    def test_minimal_object(self):
        out = drive({"type": "object", "properties": {"a": {"type": "string"}},
                     "required": ["a"]}, None)
        data = json.loads(out)
        assert isinstance(data["a"], str) and len(data["a"]) >= 1

    def test_enum_forced(self):
        out = drive({"enum": ["low", "medium", "high"]}, lambda a: a[-1])
        assert json.loads(out) in ("low", "medium", "high")

    def test_integer(self):
        out = drive({"type": "integer"}, lambda a: a[0])
        assert isinstance(json.loads(out), int)

    def test_number_with_dot(self):
        def chooser(allowed):
            if 0x2E in allowed:
                return 0x2E
            return allowed[0]

        out = drive({"type": "number"}, chooser)
        assert isinstance(json.loads(out), float)

    @pytest.mark.parametrize("schema,lo,hi", [
        ({"type": "number", "minimum": 0, "maximum": 1}, 0.0, 1.0),
        ({"type": "integer", "minimum": 1, "maximum": 5}, 1, 5),
        ({"type": "integer", "minimum": 100, "maximum": 200}, 100, 200),
        ({"type": "number", "minimum": 0.5, "maximum": 2.5}, 0.5, 2.5),
        ({"type": "integer", "maximum": 9}, float("-inf"), 9),
        ({"type": "integer", "minimum": 42}, 42, float("inf")),
    ])
    def test_number_bounds_enforced_by_construction(self, schema, lo, hi):
        """Digit-level min/max enforcement: ANY choice sequence the FSM
        allows must decode inside the bounds (advisor finding: bounds were
        stored but never enforced)."""
        for seed in range(40):
            out = drive(schema, random_chooser(seed))
            val = json.loads(out)
            assert lo <= val <= hi, (schema, out)

    def test_number_bounds_prefix_trap(self):
        """Prefix '5' can never reach [100, 200]: the first digit must be
        restricted to 1 or 2 (a contiguous-interval check would miss it)."""
        fsm = JsonFsm({"type": "integer", "minimum": 100, "maximum": 200})
        allowed = {chr(b) for b in fsm.allowed_bytes()}
        assert allowed == {"1", "2"}

    def test_unsatisfiable_bounds_fall_back_unfiltered(self):
        # min > max: keep the FSM alive (downstream clamping still applies)
        out = drive({"type": "integer", "minimum": 9, "maximum": 1},
                    lambda a: a[0])
        json.loads(out)   # still valid JSON

    def test_boolean(self):
        out = drive({"type": "boolean"}, lambda a: a[0])
        assert json.loads(out) in (True, False)

    def test_array_bounds(self):
        schema = {"type": "array", "items": {"type": "string"}, "minItems": 1, "maxItems": 3}
        # always choose to continue when possible
        def greedy(allowed):
            if 0x2C in allowed:
                return 0x2C
            if 0x22 in allowed:
                return 0x22
            return allowed[0]

        out = drive(schema, greedy)
        data = json.loads(out)
        assert 1 <= len(data) <= 3

    def test_string_max_length_enforced(self):
        schema = {"type": "string", "maxLength": 5}
        # never voluntarily close
        def never_close(allowed):
            choices = [b for b in allowed if b != 0x22]
            return choices[0] if choices else allowed[0]

        out = drive(schema, never_close)
        assert len(json.loads(out)) <= 5


@pytest.mark.parametrize("kind", sorted(PROMPT_SCHEMAS.keys()))
@pytest.mark.parametrize("seed", [0, 1, 2])
def test_all_prompt_schemas_random_choices(kind, seed):
    schema = PROMPT_SCHEMAS[kind]
    out = drive(schema, random_chooser(seed))
    data = json.loads(out)  # must always be valid JSON
    assert isinstance(data, dict)
    # and must pass the corresponding llm_parser validator
    parser = {
        "triage": parse_triage_response,
        "generateHypotheses": parse_hypothesis_generation,
        "evaluateEvidence": parse_evidence_evaluation,
        "generateConclusion": parse_conclusion,
    }.get(kind)
    if parser is not None:
        parsed = parser(out)
        assert parsed


def test_fsm_step_interface():
    fsm = JsonFsm({"type": "object", "properties": {"x": {"enum": ["a", "b"]}},
                   "required": ["x"]})
    emitted = bytearray()
    while not fsm.done:
        allowed = fsm.allowed_bytes()
        if not allowed:
            break
        b = allowed[0]
        fsm.advance(b)
        if b != NUMBER_CLOSE_SENTINEL:
            emitted.append(b)
    data = json.loads(emitted.decode())
    assert data["x"] in ("a", "b")
    assert fsm.done


class TestArrayNumberBounds:
    """Regression: the array frame chooses a number item's FIRST digit —
    it must honor the item schema's digit-level bounds (found by the
    bounds-aware fuzz; before the fix `{"maximum": 0}` items could open
    with '1' and the unsatisfiable-fallback then let anything through)."""

    def _walk_greedy(self, schema, prefer=b"975310"):
        fsm = JsonFsm(schema)
        out = bytearray()
        for _ in range(4000):
            if fsm.done:
                break
            allowed = fsm.allowed_bytes()
            if not allowed:
                break
            pick = None
            for p in prefer:  # adversarial: biggest digits first
                if p in allowed:
                    pick = p
                    break
            if pick is None:
                pick = sorted(allowed)[0]
            fsm.advance(pick)
            if pick != 0:
                out.append(pick)
        return json.loads(out.decode())

    def test_array_item_maximum(self):
        schema = {"type": "array",
                  "items": {"type": "integer", "minimum": 0, "maximum": 0},
                  "minItems": 2, "maxItems": 2}
        vals = self._walk_greedy(schema)
        assert vals == [0, 0]

    def test_array_item_minimum(self):
        schema = {"type": "array",
                  "items": {"type": "integer", "minimum": 7, "maximum": 9},
                  "minItems": 3, "maxItems": 3}
        for v in self._walk_greedy(schema):
            assert 7 <= v <= 9

    def test_array_float_bounds(self):
        schema = {"type": "array",
                  "items": {"type": "number", "minimum": 1.0, "maximum": 2.0},
                  "minItems": 2, "maxItems": 2}
        for v in self._walk_greedy(schema):
            assert 1.0 <= v <= 2.0
