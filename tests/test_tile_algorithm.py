"""CPU simulation of the GPU tile algorithm (kernels.hip k_partition +
k_merge_tiles ownership rules), validated against the oracle on randomized
shapes with SMALL tile sizes so groups span tile boundaries constantly.

This pins the algorithm — partition by (key, run) with run-order ties,
per-run +1 extras, previous-tile continuation skip, segmented winner per
group, wrapper/drop rules — independently of the LDS mechanics."""

import numpy as np
import pytest

from oracle import merge_dedup
from tests.util import random_runs


def spec_cuts(runs, D):
    """Partition spec: counts per run among the first D elements in
    (key, run) total order, ties taken in run order."""
    keys = [r["key"] for r in runs]
    # kernel bisection equivalent: smallest v with cnt_le(v) >= D
    if D == 0:
        return [0] * len(runs)
    total = sum(len(k) for k in keys)
    if D >= total:
        return [len(k) for k in keys]
    allk = np.concatenate(keys)
    runid = np.concatenate([np.full(len(k), i) for i, k in enumerate(keys)])
    order = np.lexsort((runid, allk))
    vstar = allk[order[D - 1]]  # key of the D-th smallest (1-indexed)
    cuts = []
    base = 0
    has = []
    for k in keys:
        lb = int(np.searchsorted(k, vstar, side="left"))
        cuts.append(lb)
        base += lb
        has.append(lb < len(k) and k[lb] == vstar)
    t = D - base
    assert t >= 0
    for r in range(len(keys)):
        if t > 0 and has[r]:
            cuts[r] += 1
            t -= 1
    assert t == 0
    return cuts


def simulate_tiles(runs, tile_rows, drop_delete=True, ignore_delete=False):
    k = len(runs)
    lens = [len(r["key"]) for r in runs]
    total = sum(lens)
    n_tiles = max((total + tile_rows - 1) // tile_rows, 1)
    out = []
    bounds = [spec_cuts(runs, min(b * tile_rows, total))
              for b in range(n_tiles + 1)]
    for tile in range(n_tiles):
        c0, c1 = bounds[tile], bounds[tile + 1]
        # extended segments: + 1 extra element per run where available
        elems = []  # (key, runid, row, seq, kind)
        mreal = 0
        for r in range(k):
            ext = 1 if c1[r] < lens[r] else 0
            mreal += c1[r] - c0[r]
            for row in range(c0[r], c1[r] + ext):
                elems.append((int(runs[r]["key"][row]), r, row,
                              int(runs[r]["seq"][row]),
                              int(runs[r]["kind"][row])))
        if mreal == 0:
            continue
        elems.sort(key=lambda e: (e[0], e[1]))  # stable (key, run)
        # predecessor key
        pred = None
        for r in range(k):
            if c0[r] > 0:
                kk = int(runs[r]["key"][c0[r] - 1])
                pred = kk if pred is None else max(pred, kk)
        # heads
        M = len(elems)
        heads = []
        for i in range(M):
            h = (i == 0) or (elems[i][0] != elems[i - 1][0])
            if pred is not None and elems[i][0] == pred:
                h = False
            heads.append(h)
        # groups from owned heads
        i = 0
        while i < M:
            if not heads[i] or i >= mreal:
                i += 1
                continue
            j = i
            while j + 1 < M and not heads[j + 1]:
                j += 1
            group = elems[i:j + 1]
            # winner: max (eligible, seq, isAdd)
            def isadd(kind):
                return kind in (0, 2)
            def score(e):
                elig = (not ignore_delete) or isadd(e[4])
                return (elig, e[3], isadd(e[4]))
            win = max(group, key=score)
            elig_w = (not ignore_delete) or isadd(win[4])
            emit = True
            if not elig_w and len(group) > 1:
                emit = False
            if drop_delete and not isadd(win[4]):
                emit = False
            if emit:
                out.append((win[1], win[2]))
            i = j + 1
    return out


@pytest.mark.parametrize("tile_rows", [16, 64, 256])
def test_tile_sim_vs_oracle(tile_rows):
    rng = np.random.default_rng(1000 + tile_rows)
    for trial in range(60):
        k = int(rng.integers(1, 10))
        runs = random_runs(rng, k, 300, int(rng.integers(50, 500)),
                           delete_p=0.25)
        if sum(len(r["key"]) for r in runs) == 0:
            continue
        for dd in (True, False):
            for ig in (True, False):
                got = simulate_tiles(runs, tile_rows, dd, ig)
                er, ew = merge_dedup(runs, ignore_delete=ig, drop_delete=dd)
                exp = list(zip(er.tolist(), ew.tolist()))
                assert got == exp, (trial, k, dd, ig, len(got), len(exp))


def test_tile_sim_massive_collisions():
    # nearly every key collides across all runs; groups span boundaries
    rng = np.random.default_rng(7)
    for trial in range(20):
        k = int(rng.integers(2, 9))
        n = 200
        runs = []
        seqs = rng.permutation(k * n).astype(np.int64)
        for r in range(k):
            keys = np.sort(rng.choice(n + 20, n, replace=False)).astype(np.int64)
            runs.append({"key": keys, "seq": seqs[r * n:(r + 1) * n],
                         "kind": rng.choice([0, 3], n, p=[.8, .2]).astype(np.int8)})
        got = simulate_tiles(runs, 32)
        er, ew = merge_dedup(runs)
        assert got == list(zip(er.tolist(), ew.tolist())), trial
