"""Python mirror of the reader surface over the C-ABI (ctypes).

This is plumbing for tests/bench — the host side of the product is the C++
in libpaimon_hip.so (the reference's host side is compiled (Java) code, so
ours is C++ behind a C ABI; INTEGRATION.md shows the JNI binding a Java
deployment would use instead of this module).

Mirrors SplitRead<KeyValue>.createReader (operation/SplitRead.java:39-63) /
RecordReader (paimon-common/.../reader/RecordReader.java:40-72): a
MergeReadPlan yields one batch per section; batches reuse buffers
(releaseBatch contract) — copy out if you keep them.

The product path FAILS LOUDLY when the HIP library or a GPU is missing:
there is no CPU fallback here (the CPU restatement lives in oracle/ and is
test infrastructure only).
"""

import ctypes
import json
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_HERE, "libpaimon_hip.so")

_DT_NP = {1: np.int8, 2: np.int16, 3: np.int32, 4: np.int64,
          5: np.float32, 6: np.float64,
          7: np.int32}  # PMH_DT_STRING: global-dictionary ids


class _Col(ctypes.Structure):
    _fields_ = [("name", ctypes.c_char_p), ("dtype", ctypes.c_int32),
                ("data", ctypes.c_void_p), ("valid", ctypes.c_void_p),
                ("dict_data", ctypes.c_void_p),
                ("dict_offsets", ctypes.POINTER(ctypes.c_int32)),
                ("dict_len", ctypes.c_int32),
                ("precision", ctypes.c_int32), ("scale", ctypes.c_int32)]


class _Batch(ctypes.Structure):
    _fields_ = [("n_rows", ctypes.c_int64), ("n_cols", ctypes.c_int32),
                ("device", ctypes.c_int32), ("cols", ctypes.POINTER(_Col))]


class _Stats(ctypes.Structure):
    _fields_ = [("rows_in", ctypes.c_int64), ("rows_out", ctypes.c_int64),
                ("hbm_bytes_algo", ctypes.c_int64),
                ("decode_ms", ctypes.c_double),
                ("partition_ms", ctypes.c_double),
                ("merge_ms", ctypes.c_double), ("scan_ms", ctypes.c_double),
                ("emit_ms", ctypes.c_double),
                ("total_device_ms", ctypes.c_double),
                ("h2d_ms", ctypes.c_double),
                ("path_mode", ctypes.c_int64),
                ("gpu_zstd_pages", ctypes.c_int64)]


_lib = None


def load_lib(required=True):
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(LIB_PATH):
        if required:
            raise RuntimeError(
                f"libpaimon_hip.so missing at {LIB_PATH} — build it with "
                "`make -C paimon_amd/csrc` (no CPU fallback exists; the HIP "
                "path is the product).")
        return None
    lib = ctypes.CDLL(LIB_PATH)
    lib.pmh_open_session.restype = ctypes.c_void_p
    lib.pmh_open_session.argtypes = [ctypes.c_int]
    lib.pmh_close_session.argtypes = [ctypes.c_void_p]
    lib.pmh_plan_create.restype = ctypes.c_void_p
    lib.pmh_plan_create.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
    lib.pmh_read_next.restype = ctypes.c_int64
    lib.pmh_read_next.argtypes = [ctypes.c_void_p, ctypes.POINTER(_Batch)]
    lib.pmh_changelog_next.restype = ctypes.c_int64
    lib.pmh_changelog_next.argtypes = [ctypes.c_void_p,
                                       ctypes.POINTER(_Batch)]
    lib.pmh_plan_close.argtypes = [ctypes.c_void_p]
    lib.pmh_plan_reset.argtypes = [ctypes.c_void_p]
    lib.pmh_plan_reset.restype = ctypes.c_int
    lib.pmh_stats_get.argtypes = [ctypes.c_void_p, ctypes.POINTER(_Stats)]
    lib.pmh_last_error.restype = ctypes.c_char_p
    lib.pmh_debug_footer_json.restype = ctypes.c_void_p
    lib.pmh_debug_footer_json.argtypes = [ctypes.c_char_p]
    lib.pmh_free_string.argtypes = [ctypes.c_void_p]
    lib.pmh_write_parquet.restype = ctypes.c_int
    lib.pmh_write_parquet.argtypes = [
        ctypes.POINTER(_Col), ctypes.c_int32, ctypes.c_int64,
        ctypes.c_char_p, ctypes.c_int64, ctypes.c_int64, ctypes.c_char_p]
    lib.pmh_debug_parse_dv.restype = ctypes.c_int64
    lib.pmh_debug_parse_dv.argtypes = [
        ctypes.c_char_p, ctypes.c_int64, ctypes.c_int64,
        ctypes.POINTER(ctypes.c_int64), ctypes.c_int64]
    lib.pmh_debug_interval_partition.restype = ctypes.c_int
    lib.pmh_debug_interval_partition.argtypes = [
        ctypes.c_int, ctypes.POINTER(ctypes.c_int64),
        ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int32),
        ctypes.POINTER(ctypes.c_int32)]
    _lib = lib
    return lib


def last_error():
    return load_lib().pmh_last_error().decode()


def debug_footer(path):
    lib = load_lib()
    p = lib.pmh_debug_footer_json(path.encode())
    if not p:
        raise RuntimeError(last_error())
    s = ctypes.string_at(p).decode()
    lib.pmh_free_string(p)
    return json.loads(s)


_NP_DT = {np.dtype(np.int8): 1, np.dtype(np.int16): 2,
          np.dtype(np.int32): 3, np.dtype(np.int64): 4,
          np.dtype(np.float32): 5, np.dtype(np.float64): 6}


def write_parquet(path, columns, row_group_rows=0, page_rows=0,
                  compression="NONE", dicts=None, decimals=None):
    """Write a Parquet v1 data file via the native writer
    (pmh_write_parquet) — the compaction write-back path. `columns` is an
    ordered list of (name, values[, valid]) with numpy arrays; valid
    None/omitted = REQUIRED column.

    dicts: {name: sequence of bytes/str} — the column's values are int32
    ids into that dictionary; written as a BYTE_ARRAY column with a
    dictionary page + RLE_DICTIONARY data pages (the reference writer's
    default for strings).
    decimals: {name: (precision, scale)} — int32/int64 unscaled values
    written with the DECIMAL annotation (INT32 p<=9 / INT64 p<=18,
    ParquetSchemaConverter.java:153-171)."""
    lib = load_lib()
    dicts = dicts or {}
    decimals = decimals or {}
    n_rows = len(columns[0][1]) if columns else 0
    arrs = []  # keep contiguous buffers alive
    cols = (_Col * len(columns))()
    for i, col in enumerate(columns):
        name, vals = col[0], np.ascontiguousarray(col[1])
        valid = col[2] if len(col) > 2 else None
        if vals.dtype not in _NP_DT:
            raise ValueError(f"unsupported dtype {vals.dtype} for {name}")
        if len(vals) != n_rows:
            raise ValueError("ragged columns")
        arrs.append(vals)
        cols[i].name = name.encode()
        cols[i].dtype = _NP_DT[vals.dtype]
        cols[i].data = vals.ctypes.data_as(ctypes.c_void_p)
        if name in dicts:
            if vals.dtype != np.int32:
                raise ValueError(f"{name}: dictionary ids must be int32")
            entries = [e.encode() if isinstance(e, str) else bytes(e)
                       for e in dicts[name]]
            blob = b"".join(entries)
            offs = np.zeros(len(entries) + 1, dtype=np.int32)
            np.cumsum([len(e) for e in entries], out=offs[1:])
            blob_arr = np.frombuffer(blob, dtype=np.uint8).copy() if blob \
                else np.zeros(1, dtype=np.uint8)
            arrs += [blob_arr, offs]
            cols[i].dtype = 7
            cols[i].dict_data = blob_arr.ctypes.data_as(ctypes.c_void_p)
            cols[i].dict_offsets = offs.ctypes.data_as(
                ctypes.POINTER(ctypes.c_int32))
            cols[i].dict_len = len(entries)
        elif name in decimals:
            p, s = decimals[name]
            cols[i].precision = int(p)
            cols[i].scale = int(s)
        if valid is not None:
            v8 = np.ascontiguousarray(np.asarray(valid, dtype=np.uint8))
            if len(v8) != n_rows:
                raise ValueError("ragged validity")
            arrs.append(v8)
            cols[i].valid = v8.ctypes.data_as(ctypes.c_void_p)
        else:
            cols[i].valid = None
    rc = lib.pmh_write_parquet(cols, len(columns), n_rows, path.encode(),
                               row_group_rows, page_rows,
                               compression.encode())
    if rc != 0:
        raise RuntimeError(f"pmh_write_parquet: {last_error()}")


def debug_parse_dv(path, offset=0, length=0, cap=1 << 22):
    lib = load_lib()
    out = np.empty(cap, dtype=np.int64)
    n = lib.pmh_debug_parse_dv(
        path.encode(), offset, length,
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)), cap)
    if n < 0:
        raise RuntimeError(last_error())
    return out[:min(n, cap)].copy()


def debug_zstd_cpu(data, cap=None):
    """Decode one zstd frame with the scalar zstd_core.h restatement on the
    host (test entry; the GPU page decoder shares the same core)."""
    lib = load_lib()
    lib.pmh_debug_zstd_cpu.restype = ctypes.c_int64
    lib.pmh_debug_zstd_cpu.argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                       ctypes.c_void_p, ctypes.c_int64]
    data = bytes(data)
    if cap is None:
        cap = max(16, len(data) * 64)
    out = np.empty(cap, dtype=np.uint8)
    n = lib.pmh_debug_zstd_cpu(data, len(data),
                               out.ctypes.data_as(ctypes.c_void_p), cap)
    if n < 0:
        raise RuntimeError(last_error())
    return bytes(out[:n])


def debug_zstd_gpu(data, expected_size):
    """Decode one zstd frame on the GPU (k_zstd_pages); requires a device."""
    lib = load_lib()
    lib.pmh_debug_zstd_gpu.restype = ctypes.c_int64
    lib.pmh_debug_zstd_gpu.argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                       ctypes.c_void_p, ctypes.c_int64]
    data = bytes(data)
    out = np.empty(max(1, expected_size), dtype=np.uint8)
    n = lib.pmh_debug_zstd_gpu(data, len(data),
                               out.ctypes.data_as(ctypes.c_void_p),
                               expected_size)
    if n < 0:
        raise RuntimeError(last_error())
    return bytes(out[:n])


def debug_zstd_enc_cpu(data, cap=None):
    """Compress one zstd frame with the scalar zstd_core.h encoder (host).
    Tests prove the frames decode with libzstd (pyarrow)."""
    lib = load_lib()
    lib.pmh_debug_zstd_enc_cpu.restype = ctypes.c_int64
    lib.pmh_debug_zstd_enc_cpu.argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                           ctypes.c_void_p, ctypes.c_int64]
    data = bytes(data)
    if cap is None:
        cap = len(data) + (len(data) >> 8) + 1024
    out = np.empty(cap, dtype=np.uint8)
    n = lib.pmh_debug_zstd_enc_cpu(data, len(data),
                                   out.ctypes.data_as(ctypes.c_void_p), cap)
    if n < 0:
        raise RuntimeError(last_error())
    return bytes(out[:n])


def debug_zstd_enc_gpu(data, cap=None):
    """Compress one zstd frame ON THE GPU (k_zstd_compress batch of 1)."""
    lib = load_lib()
    lib.pmh_debug_zstd_enc_gpu.restype = ctypes.c_int64
    lib.pmh_debug_zstd_enc_gpu.argtypes = [ctypes.c_void_p, ctypes.c_int64,
                                           ctypes.c_void_p, ctypes.c_int64]
    data = bytes(data)
    if cap is None:
        cap = len(data) + (len(data) >> 8) + 1024
    out = np.empty(cap, dtype=np.uint8)
    n = lib.pmh_debug_zstd_enc_gpu(data, len(data),
                                   out.ctypes.data_as(ctypes.c_void_p), cap)
    if n < 0:
        raise RuntimeError(last_error())
    return bytes(out[:n])


def interval_partition(min_keys, max_keys):
    """Returns (section_id, run_id) per input file, per the IntervalPartition
    restatement in libpaimon_hip (plan.cpp)."""
    lib = load_lib()
    n = len(min_keys)
    mn = (ctypes.c_int64 * n)(*[int(x) for x in min_keys])
    mx = (ctypes.c_int64 * n)(*[int(x) for x in max_keys])
    sec = (ctypes.c_int32 * n)()
    run = (ctypes.c_int32 * n)()
    ns = lib.pmh_debug_interval_partition(n, mn, mx, sec, run)
    if ns < 0:
        raise RuntimeError(last_error())
    return list(sec), list(run), ns


class Session:
    def __init__(self, device=0):
        self.lib = load_lib()
        self.h = self.lib.pmh_open_session(device)
        if not self.h:
            raise RuntimeError(f"pmh_open_session: {last_error()}")
        self.device = device

    def close(self):
        if self.h:
            self.lib.pmh_close_session(self.h)
            self.h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class MergeReadPlan:
    """One bucket's merge-on-read plan (MergeFileSplitRead.createReader)."""

    def __init__(self, session: Session, files, key_cols, value_cols,
                 merge_engine="deduplicate", drop_delete=True,
                 ignore_delete=False, output="host", aggregations=None,
                 remove_record_on_delete=False, sequence_groups=None,
                 ignore_retract=None, sequence_fields=None,
                 changelog_producer=None, changelog_row_dedup=False,
                 max_level=None, sort_engine="loser-tree", filters=None):
        self.lib = session.lib
        desc = {
            "key_cols": key_cols,
            "value_cols": value_cols,
            "merge_engine": merge_engine,
            "drop_delete": drop_delete,
            "ignore_delete": ignore_delete,
            "remove_record_on_delete": remove_record_on_delete,
            "output": output,
            "sort_engine": sort_engine,
            "files": files,
        }
        if sequence_groups:
            # fields.<seq>.sequence-group=<members> (CoreOptions):
            # [{"sequence_fields": [...], "group_fields": [...]}]
            desc["sequence_groups"] = list(sequence_groups)
        if ignore_retract:
            # fields.<f>.ignore-retract = true (FieldIgnoreRetractAgg)
            desc["ignore_retract"] = list(ignore_retract)
        if sequence_fields:
            # sequence.field (UserDefinedSeqComparator): value columns that
            # compare before the sequence number
            desc["sequence_fields"] = list(sequence_fields)
        if aggregations:
            # fields.<name>.aggregate-function (CoreOptions FIELDS_PREFIX);
            # unnamed columns default to last_non_null_value
            desc["aggregations"] = dict(aggregations)
        if filters:
            # value-filter conjunction; pushed into SINGLE-RUN sections
            # only (MergeFileSplitRead.java:227-239) — overlapping
            # sections emit unfiltered, as the reference reader does
            desc["filters"] = list(filters)
        if changelog_producer:
            # changelog-producer = full-compaction
            # (FullChangelogMergeFunctionWrapper; max_level = num-levels - 1)
            desc["changelog_producer"] = changelog_producer
            desc["changelog_row_deduplicate"] = bool(changelog_row_dedup)
            desc["max_level"] = int(max_level if max_level is not None
                                    else -1)
        self.h = self.lib.pmh_plan_create(session.h,
                                          json.dumps(desc).encode())
        if not self.h:
            raise RuntimeError(f"pmh_plan_create: {last_error()}")
        self.output = output

    def read_next(self):
        """Returns dict name -> numpy array (host output mode), or a raw
        _Batch with device pointers (device mode); None at end of input."""
        b = _Batch()
        n = self.lib.pmh_read_next(self.h, ctypes.byref(b))
        if n < 0:
            raise RuntimeError(f"pmh_read_next: {last_error()}")
        if n == 0 and b.n_cols == 0:
            return None
        if self.output == "device":
            return b
        out = {}
        for c in range(b.n_cols):
            col = b.cols[c]
            dt = np.dtype(_DT_NP[col.dtype])
            name = col.name.decode()
            if b.n_rows and col.data:
                buf = ctypes.cast(
                    col.data,
                    ctypes.POINTER(ctypes.c_uint8 * (b.n_rows * dt.itemsize)))
                arr = np.frombuffer(bytes(buf.contents), dtype=dt)
            else:
                arr = np.empty(0, dtype=dt)
            out[name] = arr
            if col.dtype == 7 and col.dict_len > 0:
                # dictionary string column: values are global ids; expose
                # the dictionary as an object array of bytes
                offs = np.ctypeslib.as_array(col.dict_offsets,
                                             shape=(col.dict_len + 1,))
                blob = ctypes.cast(
                    col.dict_data,
                    ctypes.POINTER(ctypes.c_uint8 * int(offs[-1])))
                bb = bytes(blob.contents) if offs[-1] else b""
                out[name + "#dict"] = np.array(
                    [bb[offs[j]:offs[j + 1]] for j in range(col.dict_len)],
                    dtype=object)
            if col.valid and b.n_rows:
                vbuf = ctypes.cast(col.valid,
                                   ctypes.POINTER(ctypes.c_uint8 * b.n_rows))
                out[name + "#valid"] = np.frombuffer(
                    bytes(vbuf.contents), dtype=np.uint8).astype(bool)
        return out

    def read_changelog(self):
        """Changelog batch of the LAST read_next (changelog_producer =
        "full-compaction"): dict name -> numpy array (host output), may be
        empty. Same schema as the main batch; _VALUE_KIND carries the
        changelog RowKind (0=+I, 1=-U, 2=+U, 3=-D)."""
        b = _Batch()
        n = self.lib.pmh_changelog_next(self.h, ctypes.byref(b))
        if n < 0:
            raise RuntimeError(f"pmh_changelog_next: {last_error()}")
        if self.output == "device":
            return b
        out = {}
        for c in range(b.n_cols):
            col = b.cols[c]
            dt = np.dtype(_DT_NP[col.dtype])
            name = col.name.decode()
            if b.n_rows and col.data:
                buf = ctypes.cast(
                    col.data,
                    ctypes.POINTER(ctypes.c_uint8 * (b.n_rows * dt.itemsize)))
                arr = np.frombuffer(bytes(buf.contents), dtype=dt)
            else:
                arr = np.empty(0, dtype=dt)
            out[name] = arr
            if col.valid and b.n_rows:
                vbuf = ctypes.cast(col.valid,
                                   ctypes.POINTER(ctypes.c_uint8 * b.n_rows))
                out[name + "#valid"] = np.frombuffer(
                    bytes(vbuf.contents), dtype=np.uint8).astype(bool)
        return out

    def reset(self):
        """Rewind to the first section without restaging (bench repeats)."""
        if self.lib.pmh_plan_reset(self.h) != 0:
            raise RuntimeError(last_error())

    def stats(self):
        s = _Stats()
        if self.lib.pmh_stats_get(self.h, ctypes.byref(s)) != 0:
            raise RuntimeError(last_error())
        return {f: getattr(s, f) for f, _ in s._fields_}

    def close(self):
        if self.h:
            self.lib.pmh_plan_close(self.h)
            self.h = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def file_descs_from_metas(metas):
    out = []
    for m in metas:
        d = {"path": m["path"], "rowCount": m["rowCount"],
             "minKey": m["minKey"], "maxKey": m["maxKey"],
             "level": m.get("level", 0)}
        if m.get("deletionVector"):
            # DeletionFile {path, offset, length}
            d["deletionVector"] = m["deletionVector"]
        out.append(d)
    return out
