"""Property-based checks over the data path (hypothesis).

These sweep randomized shapes/contents through the invariants the
hand-written fixtures cannot cover exhaustively: the vote accumulator vs a
brute-force count, RKW round-trips, and the banded aligner vs full DP."""

import numpy as np
from hypothesis import given, settings, strategies as st

from roko_amd import config as C
from roko_amd.inference import StreamingVotes, accumulate_votes


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 40), st.integers(0, 2**31 - 1))
def test_votes_match_bruteforce(n_windows, seed):
    rng = np.random.default_rng(seed)
    W = C.WINDOW_COLS
    pos = np.zeros((n_windows, W, 2), dtype=np.int64)
    pos[..., 0] = rng.integers(0, 50, (n_windows, W))
    pos[..., 1] = rng.integers(0, C.MAX_INS + 1, (n_windows, W))
    preds = rng.integers(0, C.NUM_CLASSES, (n_windows, W)).astype(np.uint8)

    keys, counts = accumulate_votes(pos, preds)
    brute = {}
    for i in range(n_windows):
        for j in range(W):
            k = (int(pos[i, j, 0]) << 3) | int(pos[i, j, 1])
            brute.setdefault(k, np.zeros(C.NUM_CLASSES, np.int64))
            brute[k][preds[i, j]] += 1
    assert list(keys) == sorted(brute)
    for k, row in zip(keys, counts):
        assert np.array_equal(row, brute[int(k)])

    # streaming with a tiny chunk gives the identical table
    sv = StreamingVotes(chunk_windows=3)
    for i in range(n_windows):
        sv.add("c", pos[i], preds[i])
    k2, c2 = sv.finalize()["c"]
    assert np.array_equal(k2, keys) and np.array_equal(c2, counts)


@settings(max_examples=20, deadline=None)
@given(st.integers(1, 12), st.integers(0, 2**31 - 1), st.booleans())
def test_rkw_roundtrip_property(n_groups, seed, inference):
    import os
    import tempfile

    from roko_amd.rkdata import RkwFile, RkwWriter

    rng = np.random.default_rng(seed)
    fd, path = tempfile.mkstemp(suffix=".rkw")
    os.close(fd)
    try:
        w = RkwWriter(path, inference=inference)
        blobs = []
        for g in range(n_groups):
            n = int(rng.integers(1, 6))
            P = rng.integers(0, 1000, (n, C.WINDOW_COLS, 2)).astype(np.int32)
            X = rng.integers(0, 12, (n, C.WINDOW_ROWS, C.WINDOW_COLS)).astype(np.uint8)
            Y = None if inference else rng.integers(0, 5, (n, C.WINDOW_COLS)).astype(np.uint8)
            w.store(f"c{g}", g * 100, g * 100 + 90, P, X, Y)
            blobs.append((P, X, Y))
        w.write_contigs([("c0", "ACGT" * 10)])
        w.close()

        f = RkwFile(path)
        assert f.num_windows == sum(len(b[0]) for b in blobs)
        for gi, (P, X, Y) in enumerate(blobs):
            _, p, x, y = f.group_arrays(gi)
            assert np.array_equal(np.asarray(p), P)
            assert np.array_equal(np.asarray(x), X)
            assert (y is None) == (Y is None)
            if Y is not None:
                assert np.array_equal(np.asarray(y), Y)
    finally:
        os.unlink(path)


@settings(max_examples=20, deadline=None)
@given(st.text(alphabet="ACGT", min_size=1, max_size=60),
       st.text(alphabet="ACGT", min_size=1, max_size=60))
def test_align_stats_property(a, b):
    from roko_amd.ops import pileup_ext

    s = pileup_ext().align_stats(a, b, band=max(len(a), len(b)))
    # triangle bounds and decomposition
    assert abs(len(a) - len(b)) <= s["edit_distance"] <= max(len(a), len(b))
    assert (s["mismatches"] + s["insertions"] + s["deletions"]
            == s["edit_distance"])
    assert s["insertions"] - s["deletions"] == len(a) - len(b)
    # symmetry of the distance (ins/del swap)
    s2 = pileup_ext().align_stats(b, a, band=max(len(a), len(b)))
    assert s2["edit_distance"] == s["edit_distance"]


@given(st.text(alphabet="ACGT", min_size=1, max_size=80),
       st.text(alphabet="ACGT", min_size=1, max_size=80))
@settings(max_examples=40, deadline=None)
def test_align_stats_metric_properties(a, b):
    """Edit distance is symmetric, zero iff equal, and bounded by the
    length difference below and max length above."""
    from roko_amd.ops import pileup_ext

    px = pileup_ext()
    d_ab = px.align_stats(a, b, band=96)["edit_distance"]
    d_ba = px.align_stats(b, a, band=96)["edit_distance"]
    assert d_ab == d_ba
    assert (d_ab == 0) == (a == b)
    assert abs(len(a) - len(b)) <= d_ab <= max(len(a), len(b))
