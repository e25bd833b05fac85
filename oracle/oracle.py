"""ctypes driver for the C oracle plus independent numpy models.

The C functions are the faithful loser-tree restatement (merge_oracle.c,
citing LoserTree.java / SortMergeReaderWithLoserTree.java). The numpy models
here are the *independent* expected-output models, restating the reference's
own test models (MergeFunctionTestUtils.java:35-85): sort by (key, seq,
isAdd), group by key, reduce. Tests require C restatement == numpy model ==
pypaimon golden vectors.
"""

import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))


def lib_path():
    return os.path.join(_HERE, "libpaimon_oracle.so")


_lib = None


def _get_lib():
    global _lib
    if _lib is None:
        p = lib_path()
        if not os.path.exists(p):
            import subprocess
            subprocess.run(["make", "-C", _HERE], check=True,
                           capture_output=True)
        _lib = ctypes.CDLL(p)
        i64p = ctypes.POINTER(ctypes.c_int64)
        i8p = ctypes.POINTER(ctypes.c_int8)
        _lib.pmo_merge_order.restype = ctypes.c_int64
        _lib.pmo_merge_dedup.restype = ctypes.c_int64
        _lib.pmo_merge_dedup_count.restype = ctypes.c_int64
        _lib.pmo_merge_dedup_count_mt.restype = ctypes.c_int64
        _lib.pmo_rle_bp_decode.restype = ctypes.c_int64
        _lib.pmo_rle_bp_decode.argtypes = [
            ctypes.c_char_p, ctypes.c_int64, ctypes.c_int, ctypes.c_int64,
            ctypes.POINTER(ctypes.c_int32)]
        del i64p, i8p
    return _lib


def _ptr_arrays(runs):
    """runs: list of dicts with contiguous numpy arrays key(i64)/seq(i64)/kind(i8)."""
    n = len(runs)
    keys = (ctypes.POINTER(ctypes.c_int64) * n)()
    seqs = (ctypes.POINTER(ctypes.c_int64) * n)()
    kinds = (ctypes.POINTER(ctypes.c_int8) * n)()
    lens = (ctypes.c_int64 * n)()
    holders = []
    for i, r in enumerate(runs):
        k = np.ascontiguousarray(r["key"], dtype=np.int64)
        s = np.ascontiguousarray(r["seq"], dtype=np.int64)
        kd = np.ascontiguousarray(r["kind"], dtype=np.int8)
        holders += [k, s, kd]
        keys[i] = k.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))
        seqs[i] = s.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))
        kinds[i] = kd.ctypes.data_as(ctypes.POINTER(ctypes.c_int8))
        lens[i] = len(k)
    return keys, seqs, kinds, lens, holders


def merge_order(runs):
    """Full merged stream: returns (run_idx i32[N], row_idx i64[N], head u8[N])."""
    lib = _get_lib()
    total = sum(len(r["key"]) for r in runs)
    out_run = np.empty(total, dtype=np.int32)
    out_row = np.empty(total, dtype=np.int64)
    out_head = np.empty(total, dtype=np.uint8)
    keys, seqs, kinds, lens, hold = _ptr_arrays(runs)
    n = lib.pmo_merge_order(
        len(runs), keys, seqs, kinds, lens,
        out_run.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        out_row.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
        out_head.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
    assert n == total, (n, total)
    return out_run, out_row, out_head


def merge_dedup(runs, ignore_delete=False, drop_delete=True):
    """C restatement of the Deduplicate merge-on-read. Returns (run, row)."""
    lib = _get_lib()
    total = sum(len(r["key"]) for r in runs)
    out_run = np.empty(max(total, 1), dtype=np.int32)
    out_row = np.empty(max(total, 1), dtype=np.int64)
    keys, seqs, kinds, lens, hold = _ptr_arrays(runs)
    n = lib.pmo_merge_dedup(
        len(runs), keys, seqs, kinds, lens,
        int(ignore_delete), int(drop_delete),
        out_run.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)),
        out_row.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)))
    return out_run[:n].copy(), out_row[:n].copy()


def _kind_is_add(kind):
    return (kind == 0) | (kind == 2)


def _sorted_stream(runs):
    """Independent model: flatten and lexsort ascending (key, seq, isAdd) —
    the total merge order of SortMergeReaderWithLoserTree.java:48-75."""
    key = np.concatenate([r["key"] for r in runs])
    seq = np.concatenate([r["seq"] for r in runs])
    kind = np.concatenate([r["kind"] for r in runs])
    run = np.concatenate([np.full(len(r["key"]), i, dtype=np.int32)
                          for i, r in enumerate(runs)])
    row = np.concatenate([np.arange(len(r["key"]), dtype=np.int64)
                          for r in runs])
    order = np.lexsort((_kind_is_add(kind).astype(np.int8), seq, key))
    return key[order], seq[order], kind[order], run[order], row[order]


def merge_dedup_count_mt(runs, n_threads, ignore_delete=False,
                         drop_delete=True):
    """N-thread dedup merge COUNT (key-space sliced loser trees) — the
    BASELINE.md N-thread CPU restatement leg. Timing baseline only."""
    lib = _get_lib()
    keys, seqs, kinds, lens, hold = _ptr_arrays(runs)
    return lib.pmo_merge_dedup_count_mt(
        len(runs), keys, seqs, kinds, lens, int(ignore_delete),
        int(drop_delete), int(n_threads))


def merge_dedup_model(runs, ignore_delete=False, drop_delete=True):
    """Numpy restatement of MergeFunctionTestUtils.getExpectedForDeduplicate
    (:35-47) extended with ignore-delete and DropDeleteReader semantics.
    Returns (run, row) of surviving records in key order."""
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    if n == 0:
        return (np.empty(0, dtype=np.int32), np.empty(0, dtype=np.int64))
    head = np.empty(n, dtype=bool)
    head[0] = True
    head[1:] = key[1:] != key[:-1]
    group_id = np.cumsum(head) - 1
    n_groups = group_id[-1] + 1
    group_size = np.bincount(group_id, minlength=n_groups)
    idx = np.arange(n)
    if ignore_delete:
        eligible = _kind_is_add(kind)
    else:
        eligible = np.ones(n, dtype=bool)
    # last eligible index per group
    last = np.full(n_groups, -1, dtype=np.int64)
    np.maximum.at(last, group_id[eligible], idx[eligible])
    # singleton groups bypass the merge function (ReducerMergeFunctionWrapper)
    single_first = np.full(n_groups, -1, dtype=np.int64)
    np.maximum.at(single_first, group_id, idx)
    singles = group_size == 1
    last[singles] = single_first[singles]
    sel = last[last >= 0]
    if drop_delete:
        sel = sel[_kind_is_add(kind[sel])]
    return run[sel], row[sel]


def partial_update_rrod_model(runs, drop_delete=True):
    """PartialUpdateMergeFunction with remove-record-on-delete
    (PartialUpdateMergeFunction.java:173-180, getResult :213-221) for
    INSERT/DELETE streams: a DELETE re-initializes the row from the DELETE
    record's own value fields; adds newer than the last DELETE overlay
    non-null fields; result kind = DELETE when the last member is the
    DELETE (dropped under drop_delete); seq = last member's."""
    key, seq, kind, run, row = _sorted_stream(runs)
    assert ((kind == 0) | (kind == 2) | (kind == 3)).all(), \
        "rrod model: INSERT/UPDATE_AFTER/DELETE only"
    n = len(key)
    n_cols = len(runs[0]["values"]) if runs else 0
    out = {"key": [], "seq": [], "kind": [],
           "values": [[] for _ in range(n_cols)],
           "valid": [[] for _ in range(n_cols)]}
    i = 0
    while i < n:
        j = i
        while j + 1 < n and key[j + 1] == key[i]:
            j += 1
        members = list(range(i, j + 1))  # ascending (seq, isAdd)
        dels = [m for m in members if kind[m] == 3]
        last_del = dels[-1] if dels else None
        res_kind = 3 if kind[members[-1]] == 3 else 0
        if not (drop_delete and res_kind == 3):
            out["key"].append(key[i])
            out["seq"].append(seq[members[-1]])
            out["kind"].append(res_kind)
            for c in range(n_cols):
                val, ok = 0, False
                for m in members:
                    if last_del is not None and m < last_del:
                        continue
                    a, b = run[m], row[m]
                    valid = runs[a].get("valid")
                    mv = bool(valid[c][b]) if valid is not None else True
                    if m == last_del:
                        # initRow: the DELETE's field value, even null
                        val = runs[a]["values"][c][b] if mv else 0
                        ok = mv
                    elif kind[m] != 3 and mv:
                        val, ok = runs[a]["values"][c][b], True
                out["values"][c].append(val if ok else 0)
                out["valid"][c].append(ok)
        i = j + 1
    return {
        "key": np.array(out["key"], np.int64),
        "seq": np.array(out["seq"], np.int64),
        "kind": np.array(out["kind"], np.int8),
        "values": [np.array(v, runs[0]["values"][c].dtype)
                   for c, v in enumerate(out["values"])],
        "valid": [np.array(v, bool) for v in out["valid"]],
    }


def partial_update_model(runs, drop_delete=True):
    """Numpy model of PartialUpdateMergeFunction for INSERT-only streams
    (PartialUpdateMergeFunction.java:188-215 updateNonNullFields +
    ReducerMergeFunctionWrapper singleton bypass). For each key group in
    ascending (seq) order, each value column takes the value of the last
    record where that column is non-null; result seq = last record's seq,
    kind = INSERT. Returns dict of output columns.

    Retract records are accepted ONLY in singleton groups: the wrapper
    bypasses the merge function there (ReducerMergeFunctionWrapper.java:
    53-73), so the lone record is served as-is with its own RowKind and a
    retract result drops under drop_delete (DropDeleteReader.java:53-61).

    runs entries carry 'values' (list of arrays, col 0 = pk) and 'valid'
    (parallel list of boolean masks)."""
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    head = np.empty(n, dtype=bool)
    if n == 0:
        return {"key": key, "seq": seq, "kind": kind, "values": [], "valid": []}
    head[0] = True
    head[1:] = key[1:] != key[:-1]
    group_id = np.cumsum(head) - 1
    n_groups = group_id[-1] + 1
    group_size = np.bincount(group_id, minlength=n_groups)
    assert (_kind_is_add(kind) | (group_size[group_id] == 1)).all(), \
        "partial_update_model: retracts only in singleton groups"
    idx = np.arange(n)
    last_all = np.zeros(n_groups, dtype=np.int64)
    np.maximum.at(last_all, group_id, idx)
    # result kind: INSERT except singleton bypass (record's own kind)
    out_kind = np.zeros(n_groups, dtype=np.int8)
    singles = group_size == 1
    out_kind[singles] = kind[last_all[singles]]
    keep = np.ones(n_groups, dtype=bool)
    if drop_delete:
        keep = _kind_is_add(out_kind)
    n_cols = len(runs[0]["values"])
    out_vals, out_valid = [], []
    # merged order over the flattened records, same as _sorted_stream
    key0 = np.concatenate([r["key"] for r in runs])
    seq0 = np.concatenate([r["seq"] for r in runs])
    kind0 = np.concatenate([r["kind"] for r in runs])
    order = np.lexsort((_kind_is_add(kind0).astype(np.int8), seq0, key0))
    for c in range(n_cols):
        col = np.concatenate([r["values"][c] for r in runs])
        msk = np.concatenate([r["valid"][c] for r in runs]) if "valid" in runs[0] \
            else np.ones(len(col), dtype=bool)
        col = col[order]
        msk = msk[order]
        lastv = np.full(n_groups, -1, dtype=np.int64)
        np.maximum.at(lastv, group_id[msk], idx[msk])
        vals = np.where(lastv >= 0, col[np.clip(lastv, 0, None)], 0)
        out_vals.append(vals.astype(col.dtype)[keep])
        out_valid.append((lastv >= 0)[keep])
    return {
        "key": key[last_all][keep], "seq": seq[last_all][keep],
        "kind": out_kind[keep],
        "values": out_vals, "valid": out_valid,
    }


def aggregation_model(runs, aggs=None, drop_delete=True):
    """Model of AggregateMergeFunction (AggregateMergeFunction.java:82-125)
    for INSERT-only streams. aggs is a list of aggregate-function names, one
    per value column (default last_non_null_value, :197-203). Per key group
    in ascending (seq, isAdd) order each column folds through its
    FieldAggregator; result seq = last member's, kind = INSERT. Singleton
    groups bypass the merge function (ReducerMergeFunctionWrapper.java:53-73)
    — so retracts are accepted in singleton groups only, served as-is with
    their own RowKind (dropped under drop_delete, DropDeleteReader.java:
    53-61).

    Follows Java numerics: int sums wrap at 64 bits here (Java's addExact
    overflow check is not modelled); float/double sums accumulate in the
    column's own precision in merge order; max/min compare like
    Float.compare/Double.compare (IEEE total order)."""
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    n_cols = len(runs[0]["values"]) if runs else 0
    if aggs is None:
        aggs = ["last_non_null_value"] * n_cols
    if n == 0:
        return {"key": key, "seq": seq, "kind": kind,
                "values": [np.empty(0, r.dtype) for r in
                           (runs[0]["values"] if runs else [])],
                "valid": [np.empty(0, bool) for _ in range(n_cols)]}
    has_valid = "valid" in runs[0]

    def fold(agg, vals, msk):
        # one group's members, merge order; returns (value, valid)
        if len(vals) == 1:  # ReducerMergeFunctionWrapper bypass
            return vals[-1], bool(msk[-1])
        if agg == "last_value":
            return vals[-1], bool(msk[-1])
        if agg == "first_value":
            return vals[0], bool(msk[0])
        if agg in ("first_non_null_value", "first_not_null_value"):
            nz = np.nonzero(msk)[0]
            return (vals[nz[0]], True) if len(nz) else (vals[0], False)
        if agg == "last_non_null_value":
            nz = np.nonzero(msk)[0]
            return (vals[nz[-1]], True) if len(nz) else (vals[0], False)
        nz = np.nonzero(msk)[0]
        if len(nz) == 0:
            return vals[0], False
        v = vals[nz]
        if agg == "sum":
            if v.dtype.kind == "f":
                acc = v.dtype.type(0)
                for x in v:  # sequential, column precision — as Java does
                    acc = v.dtype.type(acc + x)
                return acc, True
            return np.sum(v.astype(np.int64)), True
        if v.dtype.kind == "f":  # total-order max/min (Float.compare)
            ib = v.astype(v.dtype).view(
                np.int32 if v.dtype.itemsize == 4 else np.int64)
            width = 8 * v.dtype.itemsize
            ordv = np.where(ib < 0, ~ib.astype(np.uint64) & ((1 << width) - 1),
                            ib.astype(np.uint64) | (1 << (width - 1)))
            pick = np.argmax(ordv) if agg == "max" else np.argmin(ordv)
        else:
            pick = np.argmax(v) if agg == "max" else np.argmin(v)
        return v[pick], True

    head = np.empty(n, dtype=bool)
    head[0] = True
    head[1:] = key[1:] != key[:-1]
    starts = np.nonzero(head)[0]
    ends = np.append(starts[1:], n)
    sizes = ends - starts
    assert (_kind_is_add(kind) | (sizes[np.cumsum(head) - 1] == 1)).all(), \
        "aggregation_model: retracts only in singleton groups"
    cols = [np.concatenate([r["values"][c] for r in runs]) for c in
            range(n_cols)]
    msks = [np.concatenate([r["valid"][c] for r in runs]) if has_valid
            else np.ones(len(cols[c]), dtype=bool) for c in range(n_cols)]
    order = np.lexsort((
        _kind_is_add(np.concatenate([r["kind"] for r in runs])).astype(np.int8),
        np.concatenate([r["seq"] for r in runs]),
        np.concatenate([r["key"] for r in runs])))
    cols = [c[order] for c in cols]
    msks = [m[order] for m in msks]
    out_vals = [np.zeros(len(starts), dtype=c.dtype) for c in cols]
    out_valid = [np.zeros(len(starts), dtype=bool) for _ in cols]
    for g, (s, e) in enumerate(zip(starts, ends)):
        for c in range(n_cols):
            v, ok = fold(aggs[c], cols[c][s:e], msks[c][s:e])
            if ok:
                out_vals[c][g] = v
            out_valid[c][g] = ok
    out_kind = np.zeros(len(starts), dtype=np.int8)
    singles = sizes == 1
    out_kind[singles] = kind[starts[singles]]
    keep = _kind_is_add(out_kind) if drop_delete \
        else np.ones(len(starts), dtype=bool)
    return {
        "key": key[ends - 1][keep], "seq": seq[ends - 1][keep],
        "kind": out_kind[keep],
        "values": [v[keep] for v in out_vals],
        "valid": [v[keep] for v in out_valid],
    }


def aggregation_rrod_model(runs, aggs=None, drop_delete=True):
    """AggregateMergeFunction with remove-record-on-delete for INSERT/DELETE
    streams (AggregateMergeFunction.java:85-91 currentDeleteRow + initRow;
    expected-model shape: MergeFunctionTestUtils.getExpectedForAggSum's
    removeRecordOndelete branch): a DELETE re-initializes the row from its
    own value and the aggregators continue from those values; result kind =
    DELETE iff the last member is the DELETE. first_* aggregators excluded
    (their initialized-state does not reset on DELETE)."""
    key, seq, kind, run, row = _sorted_stream(runs)
    assert ((kind == 0) | (kind == 2) | (kind == 3)).all()
    n = len(key)
    n_cols = len(runs[0]["values"]) if runs else 0
    if aggs is None:
        aggs = ["last_non_null_value"] * n_cols
    assert not any(a.startswith("first") for a in aggs)
    out = {"key": [], "seq": [], "kind": [],
           "values": [[] for _ in range(n_cols)],
           "valid": [[] for _ in range(n_cols)]}
    i = 0
    while i < n:
        j = i
        while j + 1 < n and key[j + 1] == key[i]:
            j += 1
        members = list(range(i, j + 1))
        dels = [m for m in members if kind[m] == 3]
        d = dels[-1] if dels else None
        res_kind = 3 if kind[members[-1]] == 3 else 0
        if len(members) == 1:  # ReducerMergeFunctionWrapper bypass
            res_kind = int(kind[members[0]] == 3) * 3
        if not (drop_delete and res_kind == 3):
            out["key"].append(key[i])
            out["seq"].append(seq[members[-1]])
            out["kind"].append(res_kind)
            for c in range(n_cols):
                def fv(m):
                    a, b = run[m], row[m]
                    valid = runs[a].get("valid")
                    mv = bool(valid[c][b]) if valid is not None else True
                    return runs[a]["values"][c][b], mv
                if len(members) == 1:
                    val, ok = fv(members[0])
                    out["values"][c].append(val if ok else 0)
                    out["valid"][c].append(ok)
                    continue
                agg = aggs[c]
                seeds = []
                if d is not None:
                    v, mv = fv(d)
                    if mv:
                        seeds.append(v)
                adds = [m for m in members
                        if kind[m] != 3 and (d is None or m > d)]
                vals = seeds + [fv(m)[0] for m in adds if fv(m)[1]]
                if agg == "last_value":
                    val, ok = fv(members[-1])
                elif agg == "last_non_null_value":
                    ok = len(vals) > 0
                    val = vals[-1] if ok else 0
                elif agg == "sum":
                    ok = len(vals) > 0
                    val = sum(int(v) for v in vals) if ok else 0
                elif agg == "max":
                    ok = len(vals) > 0
                    val = max(vals) if ok else 0
                elif agg == "min":
                    ok = len(vals) > 0
                    val = min(vals) if ok else 0
                else:
                    raise ValueError(agg)
                out["values"][c].append(val if ok else 0)
                out["valid"][c].append(ok)
        i = j + 1
    return {
        "key": np.array(out["key"], np.int64),
        "seq": np.array(out["seq"], np.int64),
        "kind": np.array(out["kind"], np.int8),
        "values": [np.array(v, runs[0]["values"][c].dtype)
                   for c, v in enumerate(out["values"])],
        "valid": [np.array(v, bool) for v in out["valid"]],
    }


def rle_bp_decode(data: bytes, bit_width: int, num_values: int):
    """C restatement of the Parquet RLE/bit-packed hybrid decoder."""
    lib = _get_lib()
    out = np.empty(num_values, dtype=np.int32)
    n = lib.pmo_rle_bp_decode(
        data, len(data), bit_width, num_values,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_int32)))
    if n != num_values:
        raise ValueError(f"rle decode overrun: {n} != {num_values}")
    return out


def orc_rlev2_decode(data: bytes, num_values: int, signed: bool = True):
    """C restatement of ORC RLEv2 integer decode (SHORT_REPEAT / DIRECT /
    PATCHED_BASE / DELTA)."""
    lib = _get_lib()
    lib.pmo_orc_rlev2_decode.restype = ctypes.c_int64
    lib.pmo_orc_rlev2_decode.argtypes = [
        ctypes.c_char_p, ctypes.c_int64, ctypes.c_int64, ctypes.c_int,
        ctypes.POINTER(ctypes.c_int64)]
    out = np.empty(num_values, dtype=np.int64)
    n = lib.pmo_orc_rlev2_decode(
        data, len(data), num_values, int(signed),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)))
    if n != num_values:
        raise ValueError(f"rlev2 decode: {n} != {num_values}")
    return out


def orc_boolrle_decode(data: bytes, num_values: int):
    lib = _get_lib()
    lib.pmo_orc_boolrle_decode.restype = ctypes.c_int64
    lib.pmo_orc_boolrle_decode.argtypes = [
        ctypes.c_char_p, ctypes.c_int64, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_uint8)]
    out = np.empty(num_values, dtype=np.uint8)
    n = lib.pmo_orc_boolrle_decode(
        data, len(data), num_values,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_uint8)))
    if n != num_values:
        raise ValueError(f"bool rle decode: {n} != {num_values}")
    return out.astype(bool)


def merge_first_row_model(runs, ignore_delete=False, drop_delete=True):
    """Numpy restatement of getExpectedForFirstRow
    (MergeFunctionTestUtils.java:135-147) + FirstRowMergeFunction semantics
    (FirstRowMergeFunction.java:32-77): FIRST eligible record per key group
    in ascending (seq, isAdd) order; retracts are ineligible under
    ignore-delete (and reject the stream otherwise — callers pass
    insert-only or ignore_delete=True); singleton groups bypass the merge
    function. Returns (run, row)."""
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    if n == 0:
        return (np.empty(0, dtype=np.int32), np.empty(0, dtype=np.int64))
    head = np.empty(n, dtype=bool)
    head[0] = True
    head[1:] = key[1:] != key[:-1]
    group_id = np.cumsum(head) - 1
    n_groups = group_id[-1] + 1
    group_size = np.bincount(group_id, minlength=n_groups)
    idx = np.arange(n)
    eligible = _kind_is_add(kind) if ignore_delete else np.ones(n, dtype=bool)
    first = np.full(n_groups, n, dtype=np.int64)
    np.minimum.at(first, group_id[eligible], idx[eligible])
    single_first = np.full(n_groups, n, dtype=np.int64)
    np.minimum.at(single_first, group_id, idx)
    singles = group_size == 1
    first[singles] = single_first[singles]
    sel = first[first < n]
    if drop_delete:
        sel = sel[_kind_is_add(kind[sel])]
    return run[sel], row[sel]


def partial_update_seqgroup_model(runs, seq_groups, drop_delete=True,
                                  ignore_delete=False):
    """Sequential port of PartialUpdateMergeFunction with sequence groups
    (updateWithSequenceGroup :219-282, retractWithSequenceGroup :301-377,
    isEmptySequenceGroup :284-299, initRow :399-407, getResult :389-397) for
    INSERT/UPDATE_AFTER/UPDATE_BEFORE/DELETE streams, no per-field
    aggregators (their retract support is a later round).

    seq_groups: list of {"sequence_fields": [col indices],
                         "group_fields": [col indices]} — indices into the
    runs' values/valid lists. Field comparator: lexicographic ascending over
    the sequence fields, nulls FIRST (codegen nullIsLast=false).
    Returns dict of output columns like partial_update_model."""
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    n_cols = len(runs[0]["values"]) if runs else 0
    col_group = {}
    for g, sg in enumerate(seq_groups):
        for c in sg["sequence_fields"]:
            col_group[c] = g
        for c in sg["group_fields"]:
            col_group[c] = g

    def get(m, c):
        a, b = run[m], row[m]
        valid = runs[a].get("valid")
        if valid is not None and not valid[c][b]:
            return None
        return runs[a]["values"][c][b]

    def tuple_of(vals):
        # None sorts FIRST: encode as (0,) vs (1, v)
        return tuple((0, 0) if v is None else (1, int(v)) for v in vals)

    out = {"key": [], "seq": [], "kind": [],
           "values": [[] for _ in range(n_cols)],
           "valid": [[] for _ in range(n_cols)]}
    i = 0
    while i < n:
        j = i
        while j + 1 < n and key[j + 1] == key[i]:
            j += 1
        members = list(range(i, j + 1))
        if len(members) == 1:  # ReducerMergeFunctionWrapper bypass
            m = members[0]
            res_kind = int(kind[m])
            if drop_delete and res_kind in (1, 3):
                i = j + 1
                continue
            out["key"].append(key[i])
            out["seq"].append(seq[m])
            out["kind"].append(res_kind)
            for c in range(n_cols):
                v = get(m, c)
                out["values"][c].append(v if v is not None else 0)
                out["valid"][c].append(v is not None)
            i = j + 1
            continue
        rowv = [None] * n_cols
        meet_insert = False
        filled = False
        latest_seq = 0
        for m in members:
            is_retract = kind[m] in (1, 3)
            if is_retract:
                if not filled:
                    rowv = [get(m, c) for c in range(n_cols)]  # initRow
                    filled = True
                if ignore_delete:
                    continue
                latest_seq = seq[m]
                # retractWithSequenceGroup
                for g, sg in enumerate(seq_groups):
                    kvt = tuple_of([get(m, c)
                                    for c in sg["sequence_fields"]])
                    if all(t[0] == 0 for t in kvt):
                        continue  # empty sequence group
                    curt = tuple_of([rowv[c]
                                     for c in sg["sequence_fields"]])
                    if kvt >= curt:
                        for c in sg["sequence_fields"]:
                            rowv[c] = get(m, c)
                        for c in sg["group_fields"]:
                            rowv[c] = None  # retract normal field
                continue
            latest_seq = seq[m]
            meet_insert = True
            filled = True
            # updateWithSequenceGroup
            handled = set()
            for g, sg in enumerate(seq_groups):
                for c in sg["sequence_fields"]:
                    handled.add(c)
                for c in sg["group_fields"]:
                    handled.add(c)
                kvt = tuple_of([get(m, c) for c in sg["sequence_fields"]])
                if all(t[0] == 0 for t in kvt):
                    continue
                curt = tuple_of([rowv[c] for c in sg["sequence_fields"]])
                if kvt >= curt:
                    for c in sg["sequence_fields"]:
                        rowv[c] = get(m, c)
                    for c in sg["group_fields"]:
                        rowv[c] = get(m, c)  # member set EVEN IF NULL
            for c in range(n_cols):
                if c in handled:
                    continue
                v = get(m, c)
                if v is not None:
                    rowv[c] = v
        res_kind = 0 if meet_insert else 3
        if drop_delete and res_kind == 3:
            i = j + 1
            continue
        out["key"].append(key[i])
        out["seq"].append(latest_seq)
        out["kind"].append(res_kind)
        for c in range(n_cols):
            ok = rowv[c] is not None
            out["values"][c].append(rowv[c] if ok else 0)
            out["valid"][c].append(ok)
        i = j + 1
    return {
        "key": np.array(out["key"], np.int64),
        "seq": np.array(out["seq"], np.int64),
        "kind": np.array(out["kind"], np.int8),
        "values": [np.array(v, runs[0]["values"][c].dtype)
                   for c, v in enumerate(out["values"])],
        "valid": [np.array(v, bool) for v in out["valid"]],
    }


def aggregation_retract_model(runs, aggs, ignore_retract=(),
                              drop_delete=True):
    """AggregateMergeFunction with RETRACT records (AggregateMergeFunction
    .add :80-101: retract -> aggregator.retract; FieldSumAgg.retract :86-110
    subtracts with null rules; FieldPrimaryKeyAgg agg == retract == input;
    FieldIgnoreRetractAgg leaves the accumulator untouched). Only reachable
    when every column's aggregator is retract-capable — matches the plan
    validation. Result kind is INSERT (getResult: currentDeleteRow only via
    removeRecordOnDelete); singleton groups bypass the merge function.
    getExpectedForAggSum's non-RROD branch (MergeFunctionTestUtils.java:
    99-110) is the executable spec for the sum case."""
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    n_cols = len(runs[0]["values"]) if runs else 0
    ign = set(ignore_retract)

    def get(m, c):
        a, b = run[m], row[m]
        valid = runs[a].get("valid")
        if valid is not None and not valid[c][b]:
            return None
        return int(runs[a]["values"][c][b]) \
            if runs[a]["values"][c].dtype.kind in "iu" \
            else runs[a]["values"][c][b]

    out = {"key": [], "seq": [], "kind": [],
           "values": [[] for _ in range(n_cols)],
           "valid": [[] for _ in range(n_cols)]}
    i = 0
    while i < n:
        j = i
        while j + 1 < n and key[j + 1] == key[i]:
            j += 1
        members = list(range(i, j + 1))
        if len(members) == 1:
            m = members[0]
            res_kind = int(kind[m])
            if drop_delete and res_kind in (1, 3):
                i = j + 1
                continue
            out["key"].append(key[i])
            out["seq"].append(seq[m])
            out["kind"].append(res_kind)
            for c in range(n_cols):
                v = get(m, c)
                out["values"][c].append(v if v is not None else 0)
                out["valid"][c].append(v is not None)
            i = j + 1
            continue
        rowv = []
        for c in range(n_cols):
            agg = aggs[c]
            acc = None
            inited = False
            for m in members:
                rt = kind[m] in (1, 3)
                v = get(m, c)
                if agg == "primary_key":
                    acc = v
                    continue
                if rt and c in ign:
                    continue
                if rt:
                    assert agg == "sum", \
                        "retract only for sum/primary_key/ignored in v1"
                    if v is None:
                        pass
                    elif acc is None:
                        acc = -v
                    else:
                        acc = acc - v
                    continue
                if agg == "sum":
                    acc = v if acc is None else (acc if v is None
                                                 else acc + v)
                elif agg == "last_value":
                    acc = v
                elif agg == "first_value":
                    if not inited:
                        acc = v
                        inited = True
                elif agg == "last_non_null_value":
                    acc = v if v is not None else acc
                elif agg in ("first_non_null_value",
                             "first_not_null_value"):
                    acc = acc if acc is not None else v
                elif agg == "max":
                    acc = v if acc is None else (
                        acc if v is None else max(acc, v))
                elif agg == "min":
                    acc = v if acc is None else (
                        acc if v is None else min(acc, v))
            rowv.append(acc)
        out["key"].append(key[i])
        out["seq"].append(seq[members[-1]])
        out["kind"].append(0)
        for c in range(n_cols):
            ok = rowv[c] is not None
            out["values"][c].append(rowv[c] if ok else 0)
            out["valid"][c].append(ok)
        i = j + 1
    return {
        "key": np.array(out["key"], np.int64),
        "seq": np.array(out["seq"], np.int64),
        "kind": np.array(out["kind"], np.int8),
        "values": [np.array(v, runs[0]["values"][c].dtype)
                   for c, v in enumerate(out["values"])],
        "valid": [np.array(v, bool) for v in out["valid"]],
    }


def merge_dedup_useq_model(runs, useq_idx, drop_delete=True,
                           ignore_delete=False, first_row=False):
    """Deduplicate / FirstRow with user-defined sequence fields
    (utils/UserDefinedSeqComparator.java:38-80; total order becomes
    ascending (key, seq fields..., seq, isAdd), nulls FIRST). useq_idx:
    indices into the runs' values lists. Returns (run, row)."""
    key0 = np.concatenate([r["key"] for r in runs])
    seq0 = np.concatenate([r["seq"] for r in runs])
    kind0 = np.concatenate([r["kind"] for r in runs])
    run0 = np.concatenate([np.full(len(r["key"]), i, dtype=np.int32)
                           for i, r in enumerate(runs)])
    row0 = np.concatenate([np.arange(len(r["key"]), dtype=np.int64)
                           for r in runs])
    sort_keys = [_kind_is_add(kind0).astype(np.int8), seq0]
    for c in reversed(useq_idx):
        vals = np.concatenate([r["values"][c] for r in runs])
        if "valid" in runs[0]:
            msk = np.concatenate([r["valid"][c] for r in runs])
        else:
            msk = np.ones(len(vals), dtype=bool)
        sort_keys.append(np.where(msk, vals, 0))
        sort_keys.append(msk.astype(np.int8))  # nulls first
    sort_keys.append(key0)
    order = np.lexsort(tuple(sort_keys))
    key = key0[order]
    kind = kind0[order]
    run = run0[order]
    row = row0[order]
    n = len(key)
    if n == 0:
        return (np.empty(0, dtype=np.int32), np.empty(0, dtype=np.int64))
    head = np.empty(n, dtype=bool)
    head[0] = True
    head[1:] = key[1:] != key[:-1]
    group_id = np.cumsum(head) - 1
    n_groups = group_id[-1] + 1
    group_size = np.bincount(group_id, minlength=n_groups)
    idx = np.arange(n)
    eligible = _kind_is_add(kind) if ignore_delete \
        else np.ones(n, dtype=bool)
    if first_row:  # FIRST eligible per group
        pick = np.full(n_groups, n, dtype=np.int64)
        np.minimum.at(pick, group_id[eligible], idx[eligible])
        pick[pick == n] = -1
    else:  # LAST eligible per group
        pick = np.full(n_groups, -1, dtype=np.int64)
        np.maximum.at(pick, group_id[eligible], idx[eligible])
    single_first = np.full(n_groups, -1, dtype=np.int64)
    np.maximum.at(single_first, group_id, idx)
    singles = group_size == 1
    pick[singles] = single_first[singles]
    sel = pick[pick >= 0]
    if drop_delete:
        sel = sel[_kind_is_add(kind[sel])]
    return run[sel], row[sel]


def full_changelog_model(runs, levels, max_level, row_dedup=False,
                         ignore_delete=False, masks=None):
    """Numpy restatement of FullChangelogMergeFunctionWrapper.java:74-130
    over DeduplicateMergeFunction (getResult decision table; pinned to
    FullChangelogMergeFunctionWrapperTestBase's vectors in
    tests/test_changelog_cpu.py).

    levels: per-run level ints; a member is the "top level kv" iff its run's
    level == max_level (at most one per key — checkState :76-78).
    row_dedup: the valueEqualiser (compares ALL value columns).

    Returns (cl_run, cl_row, cl_kind): the changelog stream in key order
    (UPDATE_BEFORE precedes UPDATE_AFTER), where cl_kind is the changelog
    RowKind and (run,row) the source record whose key/seq/values it carries,
    plus (res_run, res_row): the result stream (= merged records that are
    adds, setResultIfNotRetract ChangelogResult.java:45).
    """
    key, seq, kind, run, row = _sorted_stream(runs)
    n = len(key)
    if n == 0:
        e32 = np.empty(0, dtype=np.int32)
        e64 = np.empty(0, dtype=np.int64)
        e8 = np.empty(0, dtype=np.int8)
        return (e32, e64, e8), (e32.copy(), e64.copy())
    head = np.empty(n, dtype=bool)
    head[0] = True
    head[1:] = key[1:] != key[:-1]
    gid = np.cumsum(head) - 1
    ng = gid[-1] + 1
    size = np.bincount(gid, minlength=ng)
    idx = np.arange(n)
    lvlarr = np.asarray(levels, dtype=np.int64)[run]
    is_top = lvlarr == max_level
    tops_per_group = np.bincount(gid[is_top], minlength=ng)
    assert (tops_per_group <= 1).all(), \
        "Top level key-value already exists (checkState :76-78)"
    top_idx = np.full(ng, -1, dtype=np.int64)
    np.maximum.at(top_idx, gid[is_top], idx[is_top])
    # merged record = deduplicate result (last in (seq, isAdd) order among
    # eligible members; ignore-delete restricts eligibility)
    eligible = _kind_is_add(kind) if ignore_delete else np.ones(n, bool)
    merged_idx = np.full(ng, -1, dtype=np.int64)
    np.maximum.at(merged_idx, gid[eligible], idx[eligible])
    single_last = np.full(ng, -1, dtype=np.int64)
    np.maximum.at(single_last, gid, idx)
    singles = size == 1
    merged_idx[singles] = single_last[singles]  # singleton wrapper bypass
    has_top = top_idx >= 0
    has_merged = merged_idx >= 0
    m_add = np.zeros(ng, dtype=bool)
    m_add[has_merged] = _kind_is_add(kind[merged_idx[has_merged]])

    # per-group changelog decision
    cl_run, cl_row, cl_kind = [], [], []
    # row_dedup equality of value columns between two flat indices
    def _values_equal(ia, ib):
        ra, xa = run[ia], row[ia]
        rb, xb = run[ib], row[ib]
        for c in range(len(runs[0]["values"])):
            na = masks[ra][c][xa] if masks else False
            nb = masks[rb][c][xb] if masks else False
            if na != nb:
                return False
            if not na and runs[ra]["values"][c][xa] != \
                    runs[rb]["values"][c][xb]:
                return False
        return True

    for g in np.flatnonzero(has_top | (has_merged & m_add)):
        t, m = top_idx[g], merged_idx[g]
        if size[g] == 1:
            # wrapper not initialized: only "no top and initial is add"
            # emits INSERT (:117-119)
            if t < 0 and m_add[g]:
                cl_run.append(run[m]); cl_row.append(row[m]); cl_kind.append(0)
            continue
        if t < 0:
            if m >= 0 and m_add[g]:
                cl_run.append(run[m]); cl_row.append(row[m]); cl_kind.append(0)
        else:
            if not (m >= 0 and m_add[g]):
                cl_run.append(run[t]); cl_row.append(row[t]); cl_kind.append(3)
            elif (not row_dedup) or (not _values_equal(t, m)):
                cl_run.append(run[t]); cl_row.append(row[t]); cl_kind.append(1)
                cl_run.append(run[m]); cl_row.append(row[m]); cl_kind.append(2)
    # result stream: merged if add (ChangelogResult.setResultIfNotRetract)
    sel = merged_idx[has_merged & m_add]
    return ((np.array(cl_run, dtype=np.int32),
             np.array(cl_row, dtype=np.int64),
             np.array(cl_kind, dtype=np.int8)),
            (run[sel], row[sel]))
