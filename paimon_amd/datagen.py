"""Seeded synthetic sorted-run generator for the merge-on-read hot path.

Writes LSM sorted runs as Parquet files using Paimon's on-disk KeyValue schema
(reference: KeyValue.createKeyValueFields, paimon-core/src/main/java/org/apache/
paimon/KeyValue.java:135-143; key-field prefix per paimon-api/src/main/java/org/
apache/paimon/table/SpecialFields.java:76-83):

    _KEY_<pk> ... , _SEQUENCE_NUMBER: int64 not null, _VALUE_KIND: int8 not null,
    <value cols> ...

Run invariants (reference: SortedRun, paimon-core/.../mergetree/SortedRun.java;
MergeTreeWriter flush): each run is sorted ascending by key and contains no
duplicate keys; sequence numbers are unique across the whole bucket.

RowKind byte encoding (paimon-api/.../types/RowKind.java:35-56):
    INSERT=0, UPDATE_BEFORE=1, UPDATE_AFTER=2, DELETE=3; isAdd = {0, 2}.
"""

import json
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq

KIND_INSERT = 0
KIND_UPDATE_BEFORE = 1
KIND_UPDATE_AFTER = 2
KIND_DELETE = 3


def _key_space(total_rows: int) -> int:
    # ~50% cross-run key collision at the C2 shape: keys uniform over
    # [0, 1.5 * total_rows).
    return max(int(total_rows * 3 // 2), 16)


def _sorted_unique_keys(rng, n, space):
    """Sorted n unique keys ~uniform over [0, space). Fast path for large n:
    draw with replacement, unique, random-subset — each key has a symmetric
    inclusion probability; exact choice() for small n."""
    if n < 500_000 or n * 2 > space:
        return np.sort(rng.choice(space, size=n, replace=False)).astype(np.int64)
    over = int(n * 1.15)
    while True:
        u = np.unique(rng.integers(0, space, size=over, dtype=np.int64))
        if len(u) >= n:
            break
        over = int(over * 1.2)
    sel = np.sort(np.argpartition(rng.random(len(u)), n)[:n])
    return u[sel]


def gen_runs_dedup(n_runs: int, rows_per_run: int, n_value_cols: int = 8,
                   seed: int = 42, delete_frac: float = 0.05):
    """Generate in-memory sorted runs for the Deduplicate configs (C1/C2).

    Schema: int64 PK + n_value_cols x int32 values (value row also contains the
    pk column first, mirroring Paimon where primary-key columns are part of the
    value row). Returns a list of dicts with numpy arrays:
      {key: i64[n], seq: i64[n], kind: i8[n], values: [i64 pk, i32 x n_value_cols]}
    """
    rng = np.random.default_rng(seed)
    total = n_runs * rows_per_run
    space = _key_space(total)
    # Globally unique sequence numbers: a shuffled range, sliced per run.
    seqs_all = rng.permutation(total).astype(np.int64)
    runs = []
    for r in range(n_runs):
        keys = _sorted_unique_keys(rng, rows_per_run, space)
        seq = seqs_all[r * rows_per_run:(r + 1) * rows_per_run]
        kind = np.where(rng.random(rows_per_run) < delete_frac,
                        KIND_DELETE, KIND_INSERT).astype(np.int8)
        values = [keys.copy()]
        for _ in range(n_value_cols):
            values.append(rng.integers(-2**31, 2**31, size=rows_per_run,
                                       dtype=np.int64).astype(np.int32))
        runs.append({"key": keys, "seq": seq, "kind": kind, "values": values})
    return runs


def gen_runs_partial_update(n_runs: int, rows_per_run: int, n_value_cols: int = 20,
                            seed: int = 43, update_frac: float = 0.3,
                            update_cols: int = 6):
    """Runs for the PartialUpdate config (C3): wide int32 rows where update
    records carry a random subset of non-null columns (rest null). All records
    are INSERTs (default partial-update rejects retracts,
    PartialUpdateMergeFunction.java:170-186)."""
    rng = np.random.default_rng(seed)
    total = n_runs * rows_per_run
    space = _key_space(total)
    seqs_all = rng.permutation(total).astype(np.int64)
    runs = []
    for r in range(n_runs):
        keys = _sorted_unique_keys(rng, rows_per_run, space)
        seq = seqs_all[r * rows_per_run:(r + 1) * rows_per_run]
        kind = np.full(rows_per_run, KIND_INSERT, dtype=np.int8)
        is_update = rng.random(rows_per_run) < update_frac
        values = [keys.copy()]  # pk col, never null
        masks = [np.ones(rows_per_run, dtype=bool)]
        # update rows set a random update_cols-subset of columns: per-row
        # random threshold rank (argpartition-free approximation: each col
        # kept with the row's top-k cut via random keys)
        uc = min(update_cols, n_value_cols)
        rk = rng.random((rows_per_run, n_value_cols))
        cut = np.partition(rk, uc - 1, axis=1)[:, uc - 1:uc]
        col_set = rk <= cut
        for c in range(n_value_cols):
            vals = rng.integers(-2**31, 2**31, size=rows_per_run,
                                dtype=np.int64).astype(np.int32)
            valid = np.where(is_update, col_set[:, c], True)
            values.append(vals)
            masks.append(valid)
        runs.append({"key": keys, "seq": seq, "kind": kind,
                     "values": values, "valid": masks})
    return runs


def run_to_arrow(run, value_names=None):
    """Convert one in-memory run to an Arrow table in paimon on-disk layout."""
    n_vals = len(run["values"])
    if value_names is None:
        value_names = ["v_k"] + [f"v_c{i}" for i in range(n_vals - 1)]
    fields = [pa.field("_KEY_k", pa.from_numpy_dtype(run["key"].dtype),
                       nullable=False),
              pa.field("_SEQUENCE_NUMBER", pa.int64(), nullable=False),
              pa.field("_VALUE_KIND", pa.int8(), nullable=False)]
    cols = [pa.array(run["key"]), pa.array(run["seq"]), pa.array(run["kind"])]
    masks = run.get("valid")
    for i, v in enumerate(run["values"]):
        fields.append(pa.field(value_names[i], pa.from_numpy_dtype(v.dtype),
                               nullable=True))
        if masks is not None:
            cols.append(pa.array(v, mask=~masks[i]))
        else:
            cols.append(pa.array(v))
    return pa.Table.from_arrays(cols, schema=pa.schema(fields))


def write_runs(runs, out_dir, compression="NONE", row_group_rows=1 << 20,
               data_page_rows=1 << 16, file_format="parquet"):
    """Write runs as Parquet or ORC data files; returns list of per-file
    metadata dicts shaped like DataFileMeta (io/DataFileMeta.java:124-190):
    fileName, rowCount, minKey, maxKey, minSequenceNumber,
    maxSequenceNumber, level."""
    os.makedirs(out_dir, exist_ok=True)
    metas = []
    for r, run in enumerate(runs):
        tbl = run_to_arrow(run)
        path = os.path.join(out_dir, f"run-{r}.{file_format}")
        if file_format == "orc":
            from pyarrow import orc as pa_orc
            pa_orc.write_table(
                tbl, path,
                compression="uncompressed" if compression == "NONE"
                else compression)
        else:
            pq.write_table(
                tbl, path,
                compression=None if compression == "NONE" else compression,
                use_dictionary=False,
                data_page_version="1.0",
                write_statistics=False,
                row_group_size=row_group_rows,
                data_page_size=data_page_rows * 8,
                store_schema=False,
            )
        metas.append({
            "path": path,
            "rowCount": int(len(run["key"])),
            "minKey": int(run["key"][0]),
            "maxKey": int(run["key"][-1]),
            "minSequenceNumber": int(run["seq"].min()),
            "maxSequenceNumber": int(run["seq"].max()),
            "level": 0 if r == 0 else 1,
        })
    manifest = os.path.join(out_dir, "files.json")
    with open(manifest, "w") as f:
        json.dump(metas, f, indent=1)
    return metas


def gen_runs_c5(n_runs: int, rows_per_run: int, seed: int = 45,
                str_card: int = 1000, delete_frac: float = 0.05):
    """Runs for the C5 compaction config (BASELINE configs[4]): mixed types —
    int64 pk, 4 x int32, decimal(18,2) (unscaled int64) and a dictionary-
    encoded string of cardinality `str_card`. The string column is carried
    as int32 ids + a shared dictionary (runs[i]["str_dict"]); the decimal as
    unscaled int64 (INT64 physical per ParquetSchemaConverter.java:153-171).
    Column order: v_k(int64), v_c0..3(int32), v_dec(decimal 18,2),
    v_str(string)."""
    rng = np.random.default_rng(seed)
    total = n_runs * rows_per_run
    space = _key_space(total)
    seqs_all = rng.permutation(total).astype(np.int64)
    sdict = [f"s{i:04d}-{rng.integers(0, 1 << 30):08x}" for i in
             range(str_card)]
    runs = []
    for r in range(n_runs):
        keys = _sorted_unique_keys(rng, rows_per_run, space)
        seq = seqs_all[r * rows_per_run:(r + 1) * rows_per_run]
        kind = np.where(rng.random(rows_per_run) < delete_frac,
                        KIND_DELETE, KIND_INSERT).astype(np.int8)
        values = [keys.copy()]
        for _ in range(4):
            values.append(rng.integers(-2**31, 2**31, size=rows_per_run,
                                       dtype=np.int64).astype(np.int32))
        values.append(rng.integers(-10**12, 10**12, size=rows_per_run,
                                   dtype=np.int64))  # unscaled decimal
        values.append(rng.integers(0, str_card, size=rows_per_run,
                                   dtype=np.int32))  # string ids
        runs.append({"key": keys, "seq": seq, "kind": kind,
                     "values": values, "str_dict": sdict})
    return runs


C5_VALUE_COLS = ([{"name": "v_k", "type": "int64"}] +
                 [{"name": f"v_c{i}", "type": "int32"} for i in range(4)] +
                 [{"name": "v_dec", "type": "decimal(18,2)"},
                  {"name": "v_str", "type": "string"}])


def write_runs_c5(runs, out_dir, writer="pyarrow", compression="NONE",
                  row_group_rows=0, data_page_rows=0):
    """Write C5 runs. writer="pyarrow" pins the on-disk form against an
    independent implementation (decimal written via store_decimal_as_integer
    so the physical type is INT64, as the reference writes it; the string
    column dictionary-encoded). writer="native" uses the library's own
    writer (fast path for bench-scale data)."""
    import decimal as _dec
    os.makedirs(out_dir, exist_ok=True)
    metas = []
    names = [c["name"] for c in C5_VALUE_COLS]
    for i, r in enumerate(runs):
        path = os.path.join(out_dir, f"run-{i}.parquet")
        n = len(r["key"])
        sdict = r["str_dict"]
        if writer == "native":
            from .reader import write_parquet as native_write
            cols = [("_KEY_k", r["key"]),
                    ("_SEQUENCE_NUMBER", r["seq"]),
                    ("_VALUE_KIND", r["kind"])]
            for c, nm in enumerate(names):
                cols.append((nm, r["values"][c]))
            native_write(path, cols, compression=compression,
                         row_group_rows=row_group_rows,
                         page_rows=data_page_rows,
                         dicts={"v_str": sdict},
                         decimals={"v_dec": (18, 2)})
        else:
            fields = [pa.field("_KEY_k", pa.int64(), nullable=False),
                      pa.field("_SEQUENCE_NUMBER", pa.int64(),
                               nullable=False),
                      pa.field("_VALUE_KIND", pa.int8(), nullable=False),
                      pa.field("v_k", pa.int64(), nullable=False)]
            fields += [pa.field(f"v_c{j}", pa.int32(), nullable=False)
                       for j in range(4)]
            fields += [pa.field("v_dec", pa.decimal128(18, 2),
                                nullable=False),
                       pa.field("v_str", pa.dictionary(pa.int32(),
                                                       pa.string()),
                                nullable=False)]
            dec_vals = [
                _dec.Decimal(int(u)).scaleb(-2)
                for u in r["values"][5]]
            arrays = ([pa.array(r["key"]), pa.array(r["seq"]),
                       pa.array(r["kind"]), pa.array(r["values"][0])] +
                      [pa.array(r["values"][1 + j]) for j in range(4)] +
                      [pa.array(dec_vals, pa.decimal128(18, 2)),
                       pa.DictionaryArray.from_arrays(
                           pa.array(r["values"][6]),
                           pa.array(sdict))])
            tbl = pa.Table.from_arrays(arrays, schema=pa.schema(fields))
            kw = {}
            if row_group_rows:
                kw["row_group_size"] = row_group_rows
            if data_page_rows:
                kw["data_page_size"] = data_page_rows * 8
            pq.write_table(tbl, path,
                           compression=None if compression in ("NONE",
                                                               None)
                           else compression.lower(),
                           use_dictionary=["v_str"],
                           store_decimal_as_integer=True,
                           data_page_version="1.0", store_schema=False,
                           **kw)
        metas.append({"path": path, "rowCount": n,
                      "minKey": int(r["key"][0]),
                      "maxKey": int(r["key"][-1]), "level": 0})
    return metas
