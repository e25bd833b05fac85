"""Compaction rewrite — the second drop-in surface:
CompactRewriter.rewrite(outputLevel, dropDelete, sections) -> CompactResult
(mergetree/compact/CompactRewriter.java:29-56), as driven by
MergeTreeCompactTask.doCompact (mergetree/compact/MergeTreeCompactTask.java:
83-162).

v1 shape (SURVEY §8a "Write-back" row): the merge runs on the GPU through
the same C-ABI plan as the read path; the Parquet *encode* side runs in
libpaimon_hip's native writer (parquet_write.cpp: PLAIN v1 pages, thrift
footer) — on-GPU page encode/compress is roadmap row §8f.1. Rolling files + per-file
DataFileMeta stats follow KeyValueDataFileWriter (io/KeyValueDataFileWriter.
java:121-170: rowCount, minKey/maxKey copies, min/max sequenceNumber,
deleteRecordCount) and RollingFileWriterImpl (roll at target row count).
"""

import os

import numpy as np

from .reader import (MergeReadPlan, Session, file_descs_from_metas,
                     write_parquet)

KIND_IS_ADD = (0, 2)


def _roll_slices(n, target_rows):
    out = []
    s = 0
    while s < n:
        out.append((s, min(s + target_rows, n)))
        s += target_rows
    return out


def rewrite(session: Session, file_metas, key_cols, value_cols, out_dir,
            output_level, drop_delete=False, merge_engine="deduplicate",
            target_file_rows=20_000_000, compression="NONE",
            file_prefix="compact", schema_id=0):
    """Merge `file_metas` (one bucket's sorted runs) and write the result as
    rolling Parquet data files. Returns CompactResult-shaped dict:
    {"before": file_metas, "after": [DataFileMeta...]} with per-file stats
    per KeyValueDataFileWriter. dropDelete semantics per
    MergeTreeCompactManager.triggerCompaction: true only when rewriting to
    the top level."""
    os.makedirs(out_dir, exist_ok=True)
    plan = MergeReadPlan(session, file_descs_from_metas(file_metas), key_cols,
                         value_cols, merge_engine=merge_engine,
                         drop_delete=drop_delete, output="host")
    # key/value split driven by the plan's declared key columns — the
    # KeyValue column layout is key cols | _SEQUENCE_NUMBER | _VALUE_KIND |
    # value cols (KeyValueSerializer.java:34-99)
    key_names = [kc["name"] for kc in key_cols]
    special = set(key_names) | {"_SEQUENCE_NUMBER", "_VALUE_KIND"}
    # DECIMAL annotations round-trip from the declared read types
    # (ParquetSchemaConverter.java:153-171); dictionary strings write back
    # with the batch's global dictionary (ids + dict page)
    decimals = {}
    for vc in list(key_cols) + list(value_cols):
        t = vc.get("type", "")
        if t.startswith("decimal(") and t.endswith(")"):
            p, s = t[8:-1].split(",")
            decimals[vc["name"]] = (int(p), int(s))
    after = []
    file_idx = 0
    try:
        while True:
            batch = plan.read_next()
            if batch is None:
                break
            n = len(batch[key_names[0]])
            if n == 0:
                continue
            seq = batch["_SEQUENCE_NUMBER"]
            kind = batch["_VALUE_KIND"]
            for s, e in _roll_slices(n, target_file_rows):
                cols = [(kn, batch[kn][s:e]) for kn in key_names]
                cols += [("_SEQUENCE_NUMBER", seq[s:e]),
                         ("_VALUE_KIND", kind[s:e])]
                dicts = {}
                for name, arr in batch.items():
                    if name in special or name.endswith("#valid") \
                            or name.endswith("#dict"):
                        continue
                    valid = batch.get(name + "#valid")
                    d = batch.get(name + "#dict")
                    if d is not None:
                        dicts[name] = d
                    cols.append((name, arr[s:e],
                                 valid[s:e] if valid is not None else None))
                path = os.path.join(out_dir, f"{file_prefix}-{file_idx}.parquet")
                file_idx += 1
                write_parquet(path, cols, compression=compression,
                              dicts=dicts, decimals=decimals)
                sq = seq[s:e]
                kd = kind[s:e]
                # minKey/maxKey are the first/last merged rows' full key
                # tuples (DataFileMeta.java:124-190); scalar for the common
                # single-column key, list for composite keys
                if len(key_names) == 1:
                    min_key = int(batch[key_names[0]][s])
                    max_key = int(batch[key_names[0]][e - 1])
                else:
                    min_key = [int(batch[kn][s]) for kn in key_names]
                    max_key = [int(batch[kn][e - 1]) for kn in key_names]
                # per-column value stats (SimpleStats valueStats +
                # nullCounts) and schemaId, completing the DataFileMeta
                # surface (io/DataFileMeta.java:124-190)
                vstats = {}
                for name, arr in batch.items():
                    if name in special or name.endswith("#valid") \
                            or name.endswith("#dict"):
                        continue
                    valid = batch.get(name + "#valid")
                    sl = arr[s:e]
                    if valid is not None:
                        lv = valid[s:e]
                        nn = sl[lv]
                        nulls = int(len(sl) - len(nn))
                    else:
                        nn = sl
                        nulls = 0
                    d = batch.get(name + "#dict")
                    if d is not None:
                        # string columns: min/max over the decoded strings
                        vals = (d[nn] if len(nn) else
                                np.empty(0, dtype=object))
                        vstats[name] = {
                            "min": (vals.min().decode()
                                    if len(vals) else None),
                            "max": (vals.max().decode()
                                    if len(vals) else None),
                            "nullCount": nulls}
                    else:
                        vstats[name] = {
                            "min": (nn.min().item() if len(nn) else None),
                            "max": (nn.max().item() if len(nn) else None),
                            "nullCount": nulls}
                after.append({
                    "path": path,
                    "fileName": os.path.basename(path),
                    "fileSize": os.path.getsize(path),
                    "rowCount": int(e - s),
                    "minKey": min_key,
                    "maxKey": max_key,
                    "minSequenceNumber": int(sq.min()),
                    "maxSequenceNumber": int(sq.max()),
                    "deleteRowCount": int(np.count_nonzero(
                        ~np.isin(kd, KIND_IS_ADD))),
                    "level": int(output_level),
                    "schemaId": int(schema_id),
                    "valueStats": vstats,
                })
    finally:
        plan.close()
    return {"before": list(file_metas), "after": after}
