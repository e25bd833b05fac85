"""full_changelog_model vs FullChangelogMergeFunctionWrapperTestBase's
vectors (reference test, restated) and vs a direct sequential port of the
wrapper (FullChangelogMergeFunctionWrapper.java:74-130) on random runs."""
import numpy as np
import pytest

from oracle import full_changelog_model

MAX_LEVEL = 3
I, UB, UA, D = 0, 1, 2, 3


def _mk(cases):
    """cases: list of (key, [(seq, kind, value, level), ...]). Builds runs
    keyed by (level-slot): one run per distinct (level, occurrence)."""
    slots = {}
    for key, members in cases:
        seen = {}
        for m in members:
            lvl = m[3]
            k = (lvl, seen.get(lvl, 0))
            seen[lvl] = seen.get(lvl, 0) + 1
            slots.setdefault(k, []).append((key, m[0], m[1], m[2]))
    runs, levels = [], []
    for (lvl, _), rows in sorted(slots.items()):
        rows.sort()
        runs.append({
            "key": np.array([r[0] for r in rows], np.int64),
            "seq": np.array([r[1] for r in rows], np.int64),
            "kind": np.array([r[2] for r in rows], np.int8),
            "values": [np.array([r[3] for r in rows], np.int32)],
        })
        levels.append(lvl)
    return runs, levels


REF_CASES = [  # FullChangelogMergeFunctionWrapperTestBase INPUT_KVS
    (1, [(1, I, 1, 0)]),
    (2, [(2, D, 0, 0)]),
    (3, [(3, I, 3, MAX_LEVEL)]),
    (4, [(4, I, 3, 0), (5, I, -3, 0)]),
    (5, [(6, I, 3, 0), (7, D, 3, 0)]),
    (6, [(8, I, 3, MAX_LEVEL), (9, I, -3, 0)]),
    (7, [(10, I, 3, MAX_LEVEL), (11, D, 3, 0)]),
    (8, [(12, I, 3, MAX_LEVEL), (13, UB, 3, 0)]),
    (9, [(14, I, 3, MAX_LEVEL), (15, I, 3, 0)]),
]


def _collect(runs, cl):
    r, w, k = cl
    return [(int(runs[a]["key"][b]), int(runs[a]["seq"][b]), int(kk),
             int(runs[a]["values"][0][b]))
            for a, b, kk in zip(r, w, k)]


class TestReferenceVectors:
    def test_without_row_dedup(self):
        runs, levels = _mk(REF_CASES)
        cl, res = full_changelog_model(runs, levels, MAX_LEVEL,
                                       row_dedup=False)
        got = _collect(runs, cl)
        exp = [(1, 1, I, 1),            # case 1: INSERT
               (4, 5, I, -3),           # case 4: INSERT of merged
               (6, 8, UB, 3), (6, 9, UA, -3),    # case 6
               (7, 10, D, 3),           # case 7: DELETE(top)
               (8, 12, D, 3),           # case 8 (UB retract): DELETE(top)
               (9, 14, UB, 3), (9, 15, UA, 3)]   # case 9, no dedup
        assert got == exp, got
        res_keys = sorted(int(runs[a]["key"][b]) for a, b in zip(*res))
        assert res_keys == [1, 3, 4, 6, 9]

    def test_with_row_dedup(self):
        runs, levels = _mk(REF_CASES)
        cl, res = full_changelog_model(runs, levels, MAX_LEVEL,
                                       row_dedup=True)
        got = _collect(runs, cl)
        exp = [(1, 1, I, 1), (4, 5, I, -3),
               (6, 8, UB, 3), (6, 9, UA, -3),
               (7, 10, D, 3), (8, 12, D, 3)]  # case 9 suppressed
        assert got == exp, got

    def test_two_top_level_members_rejected(self):
        runs, levels = _mk([(1, [(1, I, 1, MAX_LEVEL)]),
                            (2, [(2, I, 1, 0)])])
        levels = [MAX_LEVEL] * len(levels)  # force duplicate top runs
        runs2 = [dict(r) for r in runs]
        # same key in two max-level runs -> checkState fires
        runs2[1]["key"] = runs2[0]["key"].copy()
        runs2[1]["seq"] = runs2[0]["seq"] + 1
        with pytest.raises(AssertionError):
            full_changelog_model(runs2, levels, MAX_LEVEL)


def _seq_port(runs, levels, max_level, row_dedup):
    """Direct sequential port of the wrapper for fuzzing."""
    recs = []
    for ri, r in enumerate(runs):
        for i in range(len(r["key"])):
            recs.append((int(r["key"][i]), int(r["seq"][i]),
                         int(r["kind"][i]), levels[ri], ri, i))
    recs.sort(key=lambda t: (t[0], t[1], t[2] in (0, 2)))
    cl, res = [], []
    gi = 0
    isadd = lambda kd: kd in (0, 2)
    val = lambda t: tuple(runs[t[4]]["values"][c][t[5]]
                          for c in range(len(runs[0]["values"])))
    while gi < len(recs):
        gj = gi
        while gj < len(recs) and recs[gj][0] == recs[gi][0]:
            gj += 1
        grp = recs[gi:gj]
        gi = gj
        top = [t for t in grp if t[3] == max_level]
        assert len(top) <= 1
        top = top[0] if top else None
        if len(grp) == 1:
            kv = grp[0]
            if top is None and isadd(kv[2]):
                cl.append((kv, 0))
            if isadd(kv[2]):
                res.append(kv)
            continue
        merged = grp[-1]  # deduplicate: last in (seq, isAdd) order
        if top is None:
            if isadd(merged[2]):
                cl.append((merged, 0))
        else:
            if not isadd(merged[2]):
                cl.append((top, 3))
            elif (not row_dedup) or val(top) != val(merged):
                cl.append((top, 1))
                cl.append((merged, 2))
        if isadd(merged[2]):
            res.append(merged)
    return cl, res


class TestFuzzVsSequentialPort:
    @pytest.mark.parametrize("row_dedup", [False, True])
    def test_random(self, row_dedup):
        rng = np.random.default_rng(97 + int(row_dedup))
        for trial in range(25):
            k = int(rng.integers(2, 6))
            levels = [0] * (k - 1) + [MAX_LEVEL]
            runs = []
            seqbase = 0
            for ri in range(k):
                nrows = int(rng.integers(1, 400))
                keys = np.sort(rng.choice(600, nrows, replace=False))
                kind = np.where(rng.random(nrows) < 0.25, D, I).astype(np.int8)
                if ri == k - 1:
                    kind[:] = I  # top level holds compaction results
                runs.append({
                    "key": keys.astype(np.int64),
                    "seq": np.arange(nrows, dtype=np.int64) + seqbase,
                    "kind": kind,
                    "values": [rng.integers(0, 3, nrows).astype(np.int32),
                               rng.integers(0, 3, nrows).astype(np.int32)],
                })
                seqbase += nrows
            cl, res = full_changelog_model(runs, levels, MAX_LEVEL,
                                           row_dedup=row_dedup)
            ecl, eres = _seq_port(runs, levels, MAX_LEVEL, row_dedup)
            got = [(int(runs[a]["key"][b]), int(runs[a]["seq"][b]), int(kk))
                   for a, b, kk in zip(*cl)]
            exp = [(t[0], t[1], kk) for t, kk in ecl]
            assert got == exp
            gres = sorted((int(runs[a]["key"][b]), int(runs[a]["seq"][b]))
                          for a, b in zip(*res))
            assert gres == sorted((t[0], t[1]) for t in eres)


class TestFuzzTopRunPlacement:
    """The top-level run's position among the runs must not matter (top
    detection is by LEVEL, not run order) — fuzz with the max-level run
    first, middle, and last, plus null-aware row_dedup."""

    @pytest.mark.parametrize("top_pos", [0, 2, 4])
    def test_top_position(self, top_pos):
        rng = np.random.default_rng(300 + top_pos)
        for trial in range(10):
            k = 5
            runs, levels = [], []
            seqbase = 0
            for ri in range(k):
                nrows = int(rng.integers(1, 300))
                keys = np.sort(rng.choice(500, nrows, replace=False))
                kind = np.where(rng.random(nrows) < 0.3, D, I
                                ).astype(np.int8)
                lvl = MAX_LEVEL if ri == top_pos else 0
                if lvl == MAX_LEVEL:
                    kind[:] = I
                    seq = np.arange(nrows, dtype=np.int64) - 10_000
                else:
                    seq = np.arange(nrows, dtype=np.int64) + seqbase
                    seqbase += nrows
                runs.append({"key": keys.astype(np.int64), "seq": seq,
                             "kind": kind,
                             "values": [rng.integers(0, 4, nrows
                                                     ).astype(np.int32)]})
                levels.append(lvl)
            cl, res = full_changelog_model(runs, levels, MAX_LEVEL,
                                           row_dedup=bool(trial % 2))
            ecl, eres = _seq_port(runs, levels, MAX_LEVEL,
                                  bool(trial % 2))
            got = [(int(runs[a]["key"][b]), int(kk))
                   for a, b, kk in zip(*cl)]
            exp = [(t[0], kk) for t, kk in ecl]
            assert got == exp, (top_pos, trial)
