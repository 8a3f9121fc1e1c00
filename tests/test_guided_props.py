"""Property-based guided-decoding checks: random byte walks through the
JSON / schema acceptor machines must only ever produce valid JSON (and
schema-conforming JSON for SchemaMachine) whenever the machine reports
completion. The machines gate GPU sampling masks, so a wrong 'allowed'
set silently corrupts guided outputs — these walks sweep the state
space far beyond the example-based tests."""

import json

import hypothesis.strategies as st
from hypothesis import given, settings

from kserve_amd.engine.guided import JsonMachine, SchemaMachine


def random_walk(machine, rng_bytes, max_len=200):
    """Drive the machine by picking, at each step, one byte the machine
    accepts (from a candidate set ordered by the random seed bytes).
    Returns the emitted byte string (possibly incomplete)."""
    out = bytearray()
    candidates = bytes(range(32, 127)) + b"\n\t "
    for seed in rng_bytes[:max_len]:
        if machine.complete and (seed & 3) == 0:
            break  # randomly stop once complete
        # try candidate bytes starting at a seed-dependent offset
        placed = False
        for i in range(len(candidates)):
            b = candidates[(seed + i) % len(candidates)]
            trial = machine.clone()
            if trial.advance(b):
                machine.advance(b)
                out.append(b)
                placed = True
                break
        if not placed:
            break  # dead end (machine complete, no continuation allowed)
    return bytes(out)


@settings(max_examples=80, deadline=None)
@given(st.binary(min_size=8, max_size=200))
def test_json_machine_walks_produce_valid_json(rng_bytes):
    m = JsonMachine()
    out = random_walk(m, rng_bytes)
    if m.complete:
        doc = json.loads(out.decode("utf-8", "strict"))
        assert isinstance(doc, dict)  # object_only mode
    # prefix property: every emitted prefix was accepted by the machine
    check = JsonMachine()
    for b in out:
        assert check.advance(b)


SCHEMAS = [
    {"type": "object",
     "properties": {"name": {"type": "string"},
                    "age": {"type": "integer"}},
     "required": ["name", "age"]},
    {"type": "object",
     "properties": {"tags": {"type": "array",
                             "items": {"type": "string"}},
                    "ok": {"type": "boolean"}},
     "required": ["tags", "ok"]},
    {"type": "object",
     "properties": {"kind": {"enum": ["a", "b", "c"]},
                    "score": {"type": "number"}},
     "required": ["kind", "score"]},
]


@settings(max_examples=80, deadline=None)
@given(st.integers(min_value=0, max_value=len(SCHEMAS) - 1),
       st.binary(min_size=8, max_size=300))
def test_schema_machine_walks_conform(schema_idx, rng_bytes):
    schema = SCHEMAS[schema_idx]
    m = SchemaMachine(schema)
    out = random_walk(m, rng_bytes, max_len=300)
    if m.complete:
        doc = json.loads(out)
        for key in schema["required"]:
            assert key in doc, (out, key)
        props = schema["properties"]
        for key, spec in props.items():
            if key not in doc:
                continue
            t = spec.get("type")
            if t == "string":
                assert isinstance(doc[key], str)
            elif t == "integer":
                assert isinstance(doc[key], int)
            elif t == "number":
                assert isinstance(doc[key], (int, float))
            elif t == "boolean":
                assert isinstance(doc[key], bool)
            elif t == "array":
                assert isinstance(doc[key], list)
            if "enum" in spec:
                assert doc[key] in spec["enum"]


# -- router condition fuzz (same property-testing batch) ------------------

json_scalars = st.one_of(
    st.none(), st.booleans(),
    st.integers(min_value=-1000, max_value=1000),
    st.floats(-1e3, 1e3, allow_nan=False),
    st.text(st.characters(blacklist_categories=("Cs",),
                          blacklist_characters=".#"), max_size=6),
)
json_docs = st.recursive(
    json_scalars,
    lambda inner: st.one_of(
        st.lists(inner, max_size=4),
        st.dictionaries(
            st.text(st.characters(whitelist_categories=("Ll",)),
                    min_size=1, max_size=5),
            inner, max_size=4),
    ),
    max_leaves=12,
)


@settings(max_examples=150, deadline=None)
@given(json_docs, st.text(max_size=20))
def test_router_condition_never_crashes(doc, condition):
    """InferenceGraph Switch/step conditions come from user manifests:
    evaluation over ARBITRARY bodies and condition strings must return a
    bool, never raise (the reference treats invalid conditions as
    non-matching)."""
    from kserve_amd.graph.router import condition_matches, gjson_get

    assert condition_matches(doc, condition) in (True, False)
    ok, _ = gjson_get(doc, condition)
    assert ok in (True, False)


@settings(max_examples=100, deadline=None)
@given(json_docs)
def test_gjson_roundtrip_paths(doc):
    """Every reachable leaf of a document is retrievable by its own
    dotted path with exists=True."""
    from kserve_amd.graph.router import gjson_get

    def walk(node, path):
        ok, val = gjson_get(doc, path)
        assert ok, path
        assert val == node or (val != val and node != node)
        if isinstance(node, dict):
            for k, v in node.items():
                walk(v, f"{path}.{k}" if path else k)
        elif isinstance(node, list):
            for i, v in enumerate(node):
                walk(v, f"{path}.{i}" if path else str(i))

    walk(doc, "")
