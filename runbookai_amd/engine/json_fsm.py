"""JSON-schema grammar FSM for byte-level constrained decoding.

Since the engine's tokens ARE bytes (tokenizer.py), a byte-level automaton
enforces a JSON schema EXACTLY during decoding: at every step the sampler
masks logits to `allowed_bytes()`, and structural bytes (braces, keys,
colons, commas) are forced — so even a random-init model emits
schema-valid JSON. This implements the constrained-sampling plan of
SURVEY.md §7 ("we own the decoder, so enforce schema at the logits
level"), replacing the reference's hope that hosted frontier models
return parseable JSON (mitigations at llm-parser.ts:215-229).

Supported schema subset (everything llm_parser.py's PROMPT_SCHEMAS use):
object {properties, required}, array {items, minItems, maxItems},
string {maxLength}, integer/number {minimum, maximum}, boolean, enum of
strings.

Design: a stack of frames; each frame knows its allowed next bytes and
consumes one byte at a time. Keys and punctuation are emitted as FORCED
single-byte choices, so the model only "chooses" inside value positions.
"""
from __future__ import annotations

from typing import Any, Optional

# printable ASCII minus '"' and '\\' (escapes are disallowed to keep the
# automaton single-byte exact; values never need them)
_STRING_BYTES = [b for b in range(0x20, 0x7F) if b not in (0x22, 0x5C)]
_DIGITS = list(range(0x30, 0x3A))
EOS_CHOICE = -1  # sentinel: the FSM is done; sampler should emit EOS/EOT


class JsonFsm:
    def __init__(self, schema: dict[str, Any], max_total_bytes: int = 8192) -> None:
        self.schema = schema
        self.max_total_bytes = max_total_bytes
        self.emitted = 0
        self._pending: list[int] = []           # forced bytes queued for emission
        self._stack: list[dict[str, Any]] = []  # frames
        self._done = False
        self._push_value(schema)

    # -- public API -------------------------------------------------------------

    def clone(self) -> "JsonFsm":
        """Cheap state snapshot (frames are flat dicts whose nested lists —
        props, options — are never mutated, so shallow frame copies
        suffice). Used by the BPE token-trie masker (engine/grammar_bpe.py)
        to explore candidate token expansions."""
        c = JsonFsm.__new__(JsonFsm)
        c.schema = self.schema
        c.max_total_bytes = self.max_total_bytes
        c.emitted = self.emitted
        c._pending = list(self._pending)
        c._stack = [dict(f) for f in self._stack]
        c._done = self._done
        return c

    @property
    def done(self) -> bool:
        return self._done and not self._pending

    def allowed_bytes(self) -> list[int]:
        """Byte ids the next token may take; [] when generation must stop."""
        if self._pending:
            return [self._pending[0]]
        if self._done:
            return []
        frame = self._stack[-1]
        return self._frame_allowed(frame)

    def string_state(self) -> Optional[tuple[int, int]]:
        """(chars_emitted, remaining_capacity) when the FSM sits inside a
        string VALUE with no forced bytes pending; None otherwise. The BPE
        masker's fast path keys on this (grammar_bpe.py)."""
        if self._pending or self._done or not self._stack:
            return None
        frame = self._stack[-1]
        if frame.get("kind") != "string":
            return None
        cap = max(0, min(frame["max"] - frame["len"],
                         self.max_total_bytes - self.emitted))
        return frame["len"], cap

    def string_capacity(self) -> int:
        """Remaining free-content bytes when the FSM sits INSIDE a string
        value (>=1 char already emitted, no forced bytes pending); 0
        otherwise. The sampler uses this to admit multi-byte word tokens —
        words are plain printable ASCII, so any word of length <= capacity
        keeps the automaton exact (advanced byte-by-byte)."""
        if self._pending or self._done or not self._stack:
            return 0
        frame = self._stack[-1]
        if frame.get("kind") != "string" or frame["len"] == 0:
            return 0
        return max(0, min(frame["max"] - frame["len"],
                          self.max_total_bytes - self.emitted))

    def advance(self, byte: int) -> None:
        """Consume the byte the sampler chose (must be in allowed_bytes())."""
        self.emitted += 1
        if self._pending:
            expected = self._pending.pop(0)
            if byte != expected:
                raise ValueError(f"FSM desync: forced {expected}, got {byte}")
            if not self._pending and not self._stack:
                self._done = True
            return
        if self._done:
            raise ValueError("FSM already done")
        self._frame_advance(self._stack[-1], byte)

    # -- frame plumbing -----------------------------------------------------------

    def _force(self, text: str) -> None:
        self._pending.extend(text.encode("utf-8"))

    def _pop_frame(self) -> None:
        self._stack.pop()
        if not self._stack and not self._pending:
            self._done = True

    def _push_value(self, schema: dict[str, Any]) -> None:
        """Push a frame for a value of the given schema; may queue forced bytes."""
        if "enum" in schema:
            options = [str(o) for o in schema["enum"]]
            self._stack.append({"kind": "enum", "options": options, "progress": ""})
            self._force('"')
            return
        t = schema.get("type", "string")
        if t == "object":
            props = list(schema.get("properties", {}).items())
            required = set(schema.get("required", [k for k, _ in props]))
            # fixed order; only required properties are emitted (optional ones
            # would need a model choice on the KEY — structure stays forced)
            emit = [(k, v) for k, v in props if k in required] or props[:1]
            self._stack.append({"kind": "object", "props": emit, "idx": 0})
            self._force("{")
            self._object_next_key()
        elif t == "array":
            self._stack.append({
                "kind": "array",
                "items": schema.get("items", {"type": "string"}),
                "min": int(schema.get("minItems", 0)),
                "max": int(schema.get("maxItems", 8)),
                "count": 0,
                "state": "start",
            })
            self._force("[")
        elif t == "string":
            self._stack.append({"kind": "string",
                                "max": int(schema.get("maxLength", 200)),
                                "len": 0, "open": False})
            self._force('"')
        elif t in ("integer", "number"):
            self._stack.append({"kind": "number", "float": t == "number",
                                "len": 0, "max_len": 10, "has_dot": False,
                                "lit": "",
                                "minimum": schema.get("minimum"),
                                "maximum": schema.get("maximum")})
        elif t == "boolean":
            self._stack.append({"kind": "enum_raw", "options": ["true", "false"],
                                "progress": ""})
        elif t == "null":
            self._force("null")
        else:  # fallback: treat as string
            self._push_value({"type": "string"})

    def _object_next_key(self) -> None:
        frame = self._stack[-1]
        assert frame["kind"] == "object"
        if frame["idx"] >= len(frame["props"]):
            self._force("}")
            self._pop_frame()
            # chain: this object may itself complete its parent (nested
            # object values) — found by schema fuzzing, tests/test_json_fsm_fuzz.py
            self._resume_parent()
            return
        key, subschema = frame["props"][frame["idx"]]
        prefix = ", " if frame["idx"] > 0 else ""
        frame["idx"] += 1
        self._force(f'{prefix}"{key}": ')
        # value frame goes on top; when it pops, control returns to the object
        self._push_value(subschema)

    # -- per-kind allowed/advance --------------------------------------------------

    def _frame_allowed(self, frame: dict[str, Any]) -> list[int]:
        kind = frame["kind"]
        if kind == "object":
            # object frames only act through forced bytes
            return []
        if kind == "string":
            # budget guard: force close when at max
            if frame["len"] >= frame["max"] or self.emitted >= self.max_total_bytes:
                return [0x22]  # '"'
            if frame["len"] == 0:
                return _STRING_BYTES  # require at least 1 char before closing
            return _STRING_BYTES + [0x22]
        if kind == "number":
            if frame["len"] == 0:
                return self._number_filter(frame, _DIGITS)
            if frame.get("dot_pending"):
                return self._number_filter(frame, _DIGITS)  # '.' needs a digit
            if frame["len"] >= frame["max_len"]:
                return [0x00]  # sentinel handled in advance: close number
            if frame.get("leading_zero") and not frame["has_dot"]:
                # JSON forbids further digits after a leading 0
                allowed = [0x00]
                if frame["float"]:
                    allowed.append(0x2E)
                return self._number_filter(frame, allowed)
            allowed = list(_DIGITS)
            if frame["float"] and not frame["has_dot"]:
                allowed.append(0x2E)  # '.'
            allowed.append(0x00)
            return self._number_filter(frame, allowed)
        if kind in ("enum", "enum_raw"):
            progress = frame["progress"]
            nexts = {ord(o[len(progress)]) for o in frame["options"]
                     if o.startswith(progress) and len(o) > len(progress)}
            return sorted(nexts)
        if kind == "array":
            if frame["state"] == "start":
                # model chooses: emit first item or close (if min allows)
                choices = [0x7B if _opens_with_brace(frame["items"]) else 0x22]
                choices = self._array_item_start_bytes(frame)
                if frame["min"] == 0:
                    choices = choices + [0x5D]  # ']'
                return sorted(set(choices))
            if frame["state"] == "between":
                choices = []
                if frame["count"] < frame["max"]:
                    choices.append(0x2C)  # ','
                if frame["count"] >= frame["min"]:
                    choices.append(0x5D)  # ']'
                return choices or [0x5D]
        return []

    def _number_filter(self, frame: dict[str, Any], allowed: list[int]) -> list[int]:
        """Enforce schema minimum/maximum at the digit level: a byte stays
        allowed only if SOME completion of the literal lies within bounds.
        Appending digits (to the integer part or the fraction) only ever
        increases the value, so completions of literal s span
        [float(s), float(s + '9'*room)] — a byte survives iff that interval
        intersects [minimum, maximum]. The close sentinel survives iff the
        literal itself is within bounds. A schema whose bounds are
        unsatisfiable within max_len keeps the unfiltered set (the parser's
        downstream clamp still applies) rather than deadlocking the FSM."""
        lo_b = frame.get("minimum")
        hi_b = frame.get("maximum")
        if lo_b is None and hi_b is None:
            return allowed
        lo_b = float("-inf") if lo_b is None else float(lo_b)
        hi_b = float("inf") if hi_b is None else float(hi_b)
        lit = frame["lit"]
        room_total = frame["max_len"]
        is_f = frame["float"]

        def viable(s: str) -> bool:
            """Can s complete (stop now or extend with digits) in bounds?
            Fractional extensions of 'X.Y' span the contiguous interval
            [v, v+ulp); INTEGER extensions of 'p' are the disjoint union of
            [p*10^k, (p+1)*10^k) per appended digit count k — each k is
            checked separately (a contiguous-interval shortcut would let
            prefix '5' claim it reaches [100, 200])."""
            room = room_total - len(s)
            if "." in s:
                if s.endswith("."):
                    if room <= 0:
                        return False
                    return (float(s + "0") <= hi_b
                            and float(s + "9" * room) >= lo_b)
                v = float(s)
                if lo_b <= v <= hi_b:
                    return True
                return room > 0 and v <= hi_b and float(s + "9" * room) >= lo_b
            p = int(s)
            if lo_b <= p <= hi_b:
                return True
            if is_f and room >= 2 and p <= hi_b and p + 1 > lo_b:
                return True   # a fraction 'p.x' lands in [p, p+1)
            if s == "0" or room <= 0:
                return False  # JSON forbids digits after a leading zero
            ten_k = 1
            for _ in range(room):
                ten_k *= 10
                lo_k = p * ten_k
                if lo_k > hi_b:
                    return False  # grows monotonically with k
                # k appended digits span [p*10^k, (p+1)*10^k - 1] as
                # integers; fractions extend that to an interval OPEN at
                # (p+1)*10^k — so the float test must be strict ('14'
                # reaches 149.99... but never 150)
                if is_f:
                    if lo_k + ten_k > lo_b:
                        return True
                elif lo_k + ten_k - 1 >= lo_b:
                    return True
            return False

        out = []
        for b in allowed:
            if b == 0x00:
                if lit and not lit.endswith(".") and lo_b <= float(lit) <= hi_b:
                    out.append(b)
            elif b == 0x2E:
                if viable(lit + "."):
                    out.append(b)
            else:  # digit
                if viable(lit + chr(b)):
                    out.append(b)
        return out or allowed

    def _array_item_start_bytes(self, frame: dict[str, Any]) -> list[int]:
        items = frame["items"]
        if "enum" in items or items.get("type", "string") == "string":
            return [0x22]
        t = items.get("type")
        if t == "object":
            return [0x7B]
        if t == "array":
            return [0x5B]
        if t in ("integer", "number"):
            # the FIRST digit of an array number item is chosen by the
            # ARRAY frame — run it through the same digit-level bound
            # filter a live number frame would apply, or an out-of-range
            # leading digit would make the bounds unsatisfiable later
            probe = {"kind": "number", "float": t == "number", "len": 0,
                     "max_len": 10, "has_dot": False, "lit": "",
                     "minimum": items.get("minimum"),
                     "maximum": items.get("maximum")}
            return self._number_filter(probe, list(_DIGITS))
        if t == "boolean":
            return [ord("t"), ord("f")]
        return [0x22]

    def _frame_advance(self, frame: dict[str, Any], byte: int) -> None:
        kind = frame["kind"]
        if kind == "string":
            if byte == 0x22:
                self._pop_frame()
                self._resume_parent()
            else:
                frame["len"] += 1
            return
        if kind == "number":
            if byte == 0x00:
                # close-number sentinel: pop and let parent continue; the
                # sampler maps this to NOT emitting a byte (see sampler)
                self._pop_frame()
                self._resume_parent()
                return
            if byte == 0x2E:
                frame["has_dot"] = True
                frame["dot_pending"] = True
            elif frame.get("dot_pending"):
                frame["dot_pending"] = False
            if frame["len"] == 0 and byte == 0x30:
                frame["leading_zero"] = True
            frame["lit"] += chr(byte)
            frame["len"] += 1
            return
        if kind in ("enum", "enum_raw"):
            frame["progress"] += chr(byte)
            exact = frame["progress"] in frame["options"]
            extendable = any(o.startswith(frame["progress"]) and len(o) > len(frame["progress"])
                             for o in frame["options"])
            if exact and not extendable:
                if kind == "enum":
                    self._force('"')
                self._pop_frame()
                self._resume_parent()
            elif exact and extendable:
                # prefix of a longer option (e.g. "low"/"lowest"): prefer exact
                if kind == "enum":
                    self._force('"')
                self._pop_frame()
                self._resume_parent()
            return
        if kind == "array":
            if frame["state"] == "start":
                if byte == 0x5D:
                    self._pop_frame()
                    self._resume_parent()
                    return
                frame["state"] = "between"
                frame["count"] = 1
                self._start_item(frame, byte)
                return
            if frame["state"] == "between":
                if byte == 0x5D:
                    self._pop_frame()
                    self._resume_parent()
                elif byte == 0x2C:
                    frame["count"] += 1
                    self._force(" ")
                    self._push_value(frame["items"])  # forced prefix lands after the space
                return
        raise ValueError(f"frame {kind} cannot advance on byte {byte}")

    def _start_item(self, frame: dict[str, Any], first_byte: int) -> None:
        """The model chose to start an array item; push the item frame and
        replay the first byte into it."""
        items = frame["items"]
        self._push_value(items)
        # _push_value may have queued forced bytes (e.g. '"' or '{'); the
        # first byte the model emitted IS that forced byte — consume it.
        if self._pending and self._pending[0] == first_byte:
            self._pending.pop(0)
            if not self._pending and frame.get("_noop"):
                pass
        else:
            # number/boolean items: no forced prefix; replay into the new frame
            self._frame_advance(self._stack[-1], first_byte)

    def _resume_parent(self) -> None:
        """After a value frame pops, hand control back to its parent."""
        if not self._stack:
            return
        parent = self._stack[-1]
        if parent["kind"] == "object":
            self._object_next_key()
        # array parents continue from their own "between" state naturally


NUMBER_CLOSE_SENTINEL = 0x00  # exposed for the sampler


def generate_minimal(schema: dict[str, Any], chooser=None, max_bytes: int = 8192) -> str:
    """Drive the FSM with a chooser(allowed)->byte (default: first allowed).
    Used in tests and as a CPU fallback 'model'."""
    fsm = JsonFsm(schema, max_total_bytes=max_bytes)
    out = bytearray()
    chooser = chooser or (lambda allowed: allowed[0])
    steps = 0
    while not fsm.done and steps < max_bytes * 2:
        steps += 1
        allowed = fsm.allowed_bytes()
        if not allowed:
            break
        byte = chooser(allowed)
        fsm.advance(byte)
        if byte != NUMBER_CLOSE_SENTINEL:
            out.append(byte)
    return out.decode("utf-8", errors="replace")


def _opens_with_brace(schema: dict[str, Any]) -> bool:
    return schema.get("type") == "object"
