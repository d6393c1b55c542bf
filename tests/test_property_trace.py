"""Property tests (hypothesis) for the real-trace CSV loader
(traces/philly.py): any structurally-valid CSV must load into a valid
JobTrace, and windows must be verbatim sorted slices."""
import numpy as np
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from rlgpuschedule_amd.traces.philly import load_trace_csv, windows_from_trace

rows = st.lists(
    st.tuples(
        st.floats(min_value=0, max_value=1e7, allow_nan=False,
                  allow_infinity=False),            # submit_time s
        st.integers(min_value=0, max_value=200),    # gpus (0 -> skipped)
        st.floats(min_value=0, max_value=1e6, allow_nan=False,
                  allow_infinity=False),            # duration s (0 -> skip)
        st.integers(min_value=1, max_value=200),    # min_gpus
    ),
    min_size=1, max_size=120,
)


@settings(max_examples=80, deadline=None)
@given(rows=rows, max_gpus=st.integers(min_value=1, max_value=64),
       oversize=st.sampled_from(["clip", "drop"]))
def test_loader_invariants(tmp_path_factory, rows, max_gpus, oversize):
    p = tmp_path_factory.mktemp("tr") / "t.csv"
    lines = ["submit_time,gpus,duration_s,min_gpus"]
    usable = 0
    for a, g, d, m in rows:
        lines.append(f"{a},{g},{d},{m}")
        if g >= 1 and d > 0 and not (oversize == "drop" and g > max_gpus):
            usable += 1
    p.write_text("\n".join(lines) + "\n")
    if usable == 0:
        with pytest.raises(ValueError):
            load_trace_csv(p, max_gpus, oversize=oversize)
        return
    tr = load_trace_csv(p, max_gpus, oversize=oversize)
    assert len(tr) == usable
    tr.validate(max_gpus)
    assert tr.arrival_us[0] == 0
    assert (np.diff(tr.arrival_us) >= 0).all()
    assert (tr.min_gpus <= tr.gpus).all()
    assert (tr.work_us >= 1).all()
    if oversize == "clip":
        assert tr.gpus.max() <= max_gpus


@settings(max_examples=30, deadline=None)
@given(num_envs=st.integers(min_value=1, max_value=9),
       jpe=st.integers(min_value=1, max_value=40),
       seed=st.integers(min_value=0, max_value=2**31 - 1))
def test_window_invariants(num_envs, jpe, seed):
    from rlgpuschedule_amd.traces.synthetic import philly_like
    tr = philly_like(40, seed=1, max_job_gpus=64)
    if jpe > len(tr):
        with pytest.raises(ValueError):
            windows_from_trace(tr, num_envs, jpe, seed)
        return
    pool, lo, hi = windows_from_trace(tr, num_envs, jpe, seed)
    assert len(pool) == num_envs * jpe
    for e in range(num_envs):
        assert hi[e] - lo[e] == jpe
        w = pool.arrival_us[lo[e]:hi[e]]
        assert (np.diff(w) >= 0).all()
        pool.validate(64, int(lo[e]), int(hi[e]))
