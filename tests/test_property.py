"""Hypothesis property tests for pure-python / native helpers.

Mirrors the reference's fuzz-by-variety style: instead of one fixed
oracle example, random shapes exercise the edge conditions (documents
shorter than a sample, rampup boundaries, padding) that fixed tests
tend to miss.
"""

import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=60, deadline=None)
@given(
    sizes=st.lists(st.integers(min_value=1, max_value=40), min_size=1,
                   max_size=30),
    seq_length=st.integers(min_value=1, max_value=17),
    num_epochs=st.integers(min_value=1, max_value=4),
)
def test_build_sample_idx_native_matches_python(sizes, seq_length,
                                                num_epochs):
    """Native C++ build_sample_idx == the pure-python reference for
    arbitrary document length distributions."""
    from megatronapp_amd.core.datasets.gpt_dataset import (
        _build_sample_idx_py)
    from megatronapp_amd.core.datasets.build_helpers import load_helpers

    sizes = np.array(sizes, dtype=np.int32)
    doc_idx = np.concatenate(
        [np.arange(len(sizes), dtype=np.int32)] * num_epochs)
    tokens_per_epoch = int(sizes.sum())
    if num_epochs * tokens_per_epoch - 1 < seq_length:
        return  # not enough tokens for a single sample
    ref = _build_sample_idx_py(sizes, doc_idx, seq_length, num_epochs,
                               tokens_per_epoch)
    got = load_helpers().build_sample_idx(
        sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch)
    assert np.array_equal(np.asarray(got), ref)


@settings(max_examples=80, deadline=None)
@given(
    mbs=st.integers(min_value=1, max_value=8),
    dp=st.integers(min_value=1, max_value=8),
    gbs_mult=st.integers(min_value=1, max_value=32),
    start_mult=st.integers(min_value=1, max_value=8),
    incr_mult=st.integers(min_value=1, max_value=4),
    ramup=st.integers(min_value=1, max_value=10_000),
    consumed=st.integers(min_value=0, max_value=20_000),
)
def test_rampup_microbatch_calculator_invariants(mbs, dp, gbs_mult,
                                                 start_mult, incr_mult,
                                                 ramup, consumed):
    """Rampup calculator invariants for arbitrary configs: the current
    global batch is always a positive multiple of mbs*dp, never exceeds
    the final size, is monotone in consumed samples, and reaches the
    final size when the ramp ends."""
    from megatronapp_amd.core.num_microbatches_calculator import (
        RampupBatchsizeNumMicroBatchesCalculator)

    unit = mbs * dp
    final = unit * max(gbs_mult, start_mult)
    start = unit * min(gbs_mult, start_mult)
    calc = RampupBatchsizeNumMicroBatchesCalculator(
        final, mbs, dp, start, incr_mult * unit, ramup)

    prev = 0
    for c in sorted({0, consumed // 2, consumed, ramup, ramup + 1}):
        calc.update(c, consistency_check=False)
        gbs = calc.get_current_global_batch_size()
        assert gbs % unit == 0 and gbs >= unit
        assert gbs <= final
        assert gbs >= prev, "global batch must not shrink during rampup"
        assert calc.get() == gbs // unit
        prev = gbs
    calc.update(ramup, consistency_check=False)
    assert calc.get_current_global_batch_size() == final


@settings(max_examples=60, deadline=None)
@given(
    weights=st.lists(st.floats(min_value=0.01, max_value=1.0),
                     min_size=1, max_size=6),
    size=st.integers(min_value=1, max_value=500),
)
def test_blending_indices_invariants(weights, size):
    """Greedy blending: per-dataset sample indices are 0..k-1 dense in
    order, every position assigned, and each realized share stays within
    one sample of its normalized weight."""
    from megatronapp_amd.core.datasets.build_helpers import load_helpers

    w = np.array(weights, dtype=np.float64)
    w = w / w.sum()
    di, dsi = load_helpers().build_blending_indices(w, size)
    di, dsi = np.asarray(di), np.asarray(dsi)
    assert di.shape == (size,) and dsi.shape == (size,)
    counts = np.zeros(len(w), dtype=np.int64)
    for d, s in zip(di, dsi):
        assert s == counts[d], "per-dataset indices must be dense in order"
        counts[d] += 1
    assert counts.sum() == size
    # realized share within 1 sample of target at the end
    for d in range(len(w)):
        assert abs(counts[d] - w[d] * size) <= len(w), (counts, w * size)
