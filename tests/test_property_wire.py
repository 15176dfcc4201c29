"""Property tests (hypothesis): the wire-facing normalizers never raise on
arbitrary input — the aggregator must survive any peer."""

import json

from hypothesis import given, settings
from hypothesis import strategies as st

from traceml_amd.steptime.repository import normalize_step_time_events
from traceml_amd.telemetry.control import parse_control
from traceml_amd.telemetry.envelope import normalize_telemetry_envelope

_json_scalars = st.one_of(
    st.none(),
    st.booleans(),
    st.integers(min_value=-(2**53), max_value=2**53),
    st.floats(allow_nan=False, allow_infinity=False),
    st.text(max_size=40),
)

_json_values = st.recursive(
    _json_scalars,
    lambda children: st.one_of(
        st.lists(children, max_size=5),
        st.dictionaries(st.text(max_size=20), children, max_size=5),
    ),
    max_leaves=25,
)


@settings(max_examples=200, deadline=None)
@given(_json_values)
def test_normalize_envelope_never_raises(payload):
    result = normalize_telemetry_envelope(payload)
    if result is not None:
        assert isinstance(result["meta"]["sampler"], str)
        assert isinstance(result["body"]["tables"], dict)
        for rows in result["body"]["tables"].values():
            assert all(isinstance(r, dict) for r in rows)


@settings(max_examples=200, deadline=None)
@given(_json_values)
def test_parse_control_never_raises(payload):
    result = parse_control(payload)
    if result is not None:
        assert isinstance(result["_traceml_control"], str)


@settings(max_examples=200, deadline=None)
@given(_json_values)
def test_normalize_events_never_raises(payload):
    raw = json.dumps(payload)
    result = normalize_step_time_events(raw)
    if result is not None:
        for cell in result.values():
            assert set(cell) == {
                "duration_ms", "cpu_ms", "gpu_ms", "n_calls", "is_gpu",
            }


@settings(max_examples=100, deadline=None)
@given(st.binary(max_size=200))
def test_normalize_events_garbage_strings(blob):
    try:
        raw = blob.decode("utf-8", errors="replace")
    except Exception:
        return
    normalize_step_time_events(raw)  # must not raise


@settings(max_examples=150, deadline=None)
@given(
    st.lists(
        st.tuples(
            st.integers(min_value=0, max_value=7),   # rank
            st.integers(min_value=1, max_value=40),  # step
            _json_values,                            # events payload
        ),
        max_size=30,
    )
)
def test_analyzer_never_raises_on_normalized_rows(raw_rows):
    from traceml_amd.steptime.analyzer import StepTimeAnalyzer
    from traceml_amd.steptime.model import StepTimeSourceRow

    rows = []
    for i, (rank, step, payload) in enumerate(raw_rows):
        events = normalize_step_time_events(json.dumps(payload))
        if events is None:
            continue
        rows.append(
            StepTimeSourceRow(
                row_id=i, global_rank=rank, step=step,
                timestamp=float(step), events=events,
            )
        )
    window = StepTimeAnalyzer().analyze(rows)
    # invariants hold whatever came in
    for values in window.ranks.values():
        st_ms = values.get("step_time_ms")
        assert st_ms is None or st_ms == st_ms  # no NaN
