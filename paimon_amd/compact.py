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
            file_prefix="compact"):
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
    after = []
    file_idx = 0
    try:
        while True:
            batch = plan.read_next()
            if batch is None:
                break
            n = len(batch["_KEY_k"])
            if n == 0:
                continue
            key = batch["_KEY_k"]
            seq = batch["_SEQUENCE_NUMBER"]
            kind = batch["_VALUE_KIND"]
            for s, e in _roll_slices(n, target_file_rows):
                cols = [("_KEY_k", key[s:e]), ("_SEQUENCE_NUMBER", seq[s:e]),
                        ("_VALUE_KIND", kind[s:e])]
                for name, arr in batch.items():
                    if name in ("_KEY_k", "_SEQUENCE_NUMBER", "_VALUE_KIND") \
                            or name.endswith("#valid"):
                        continue
                    valid = batch.get(name + "#valid")
                    cols.append((name, arr[s:e],
                                 valid[s:e] if valid is not None else None))
                path = os.path.join(out_dir, f"{file_prefix}-{file_idx}.parquet")
                file_idx += 1
                write_parquet(path, cols, compression=compression)
                ks = key[s:e]
                sq = seq[s:e]
                kd = kind[s:e]
                after.append({
                    "path": path,
                    "fileName": os.path.basename(path),
                    "fileSize": os.path.getsize(path),
                    "rowCount": int(e - s),
                    "minKey": int(ks[0]),
                    "maxKey": int(ks[-1]),
                    "minSequenceNumber": int(sq.min()),
                    "maxSequenceNumber": int(sq.max()),
                    "deleteRowCount": int(np.count_nonzero(
                        ~np.isin(kd, KIND_IS_ADD))),
                    "level": int(output_level),
                })
    finally:
        plan.close()
    return {"before": list(file_metas), "after": after}
