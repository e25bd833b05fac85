"""Drive pypaimon — the reference's own Python implementation — as a parity
pin for the oracle. CONTAINER-ONLY: requires /root/reference to exist; never
imported by -m gpu tests, smoke() or bench.py. Used by
tests gated on the reference being present and by gen_golden.py to produce
the committed fixtures under tests/golden/.

pypaimon imports after stubbing 4 pure-Python deps absent from this image
(cachetools, readerwriterlock, polars, fastavro — import-only stubs).
"""

import sys
import types

REFERENCE_PYPAIMON = "/root/reference/paimon-python"


def available():
    import os
    return os.path.isdir(REFERENCE_PYPAIMON)


_loaded = False


def _ensure_loaded():
    global _loaded
    if _loaded:
        return
    if not available():
        raise RuntimeError("pypaimon reference not available in this container")
    for name in ("cachetools", "readerwriterlock", "polars", "fastavro"):
        if name not in sys.modules:
            mod = types.ModuleType(name)
            if name == "cachetools":
                class _Cache(dict):
                    def __init__(self, maxsize=0, ttl=0, **kw):
                        super().__init__()
                mod.TTLCache = _Cache
                mod.LRUCache = _Cache
                mod.cached = lambda *a, **k: (lambda f: f)
            if name == "readerwriterlock":
                class _Lock:
                    def __init__(self, *a, **k):
                        pass

                    class _L:
                        def __enter__(self):
                            return self

                        def __exit__(self, *a):
                            return False

                        def acquire(self):
                            return True

                        def release(self):
                            pass

                    def gen_rlock(self):
                        return self._L()

                    def gen_wlock(self):
                        return self._L()
                rwlock = types.ModuleType("readerwriterlock.rwlock")
                rwlock.RWLockFair = _Lock
                rwlock.RWLockRead = _Lock
                rwlock.RWLockWrite = _Lock
                mod.rwlock = rwlock
                sys.modules["readerwriterlock.rwlock"] = rwlock
            sys.modules[name] = mod
    if REFERENCE_PYPAIMON not in sys.path:
        sys.path.insert(0, REFERENCE_PYPAIMON)
    _loaded = True


def merge_with_pypaimon(runs, merge_function_name="deduplicate",
                        drop_delete=True, n_value_cols=None):
    """Run pypaimon's SortMergeReaderWithMinHeap over in-memory runs.

    runs: list of dicts with numpy arrays key/seq/kind (+ optional 'values'
    list and 'valid' masks). Returns list of merged records as tuples
    (key, seq, kind_byte, values tuple) in merged order, after optional
    drop-delete filtering (DropDeleteReader semantics).
    """
    _ensure_loaded()
    from pypaimon.read.reader.iface.record_reader import RecordReader
    from pypaimon.read.reader.iface.record_iterator import RecordIterator
    from pypaimon.read.reader.sort_merge_reader import SortMergeReaderWithMinHeap
    from pypaimon.table.row.key_value import KeyValue
    from pypaimon.schema.data_types import AtomicType, DataField
    from pypaimon.schema.table_schema import TableSchema

    if n_value_cols is None:
        n_value_cols = len(runs[0].get("values", [])) if runs else 0

    key_arity = 1

    class MemIterator(RecordIterator):
        def __init__(self, run):
            self.run = run
            self.i = 0
            self.n = len(run["key"])

        def next(self):
            if self.i >= self.n:
                return None
            r = self.run
            i = self.i
            self.i += 1
            vals = tuple(
                (r["values"][c][i].item()
                 if ("valid" not in r or r["valid"][c][i]) else None)
                for c in range(n_value_cols))
            tup = (r["key"][i].item(), r["seq"][i].item(),
                   int(r["kind"][i])) + vals
            kv = KeyValue(key_arity, n_value_cols)
            return kv.replace(tup)

    class MemReader(RecordReader):
        def __init__(self, run):
            self.run = run
            self.done = False

        def read_batch(self):
            if self.done:
                return None
            self.done = True
            return MemIterator(self.run)

        def close(self):
            pass

    fields = [DataField(0, "k", AtomicType("BIGINT", nullable=False))]
    for c in range(n_value_cols):
        fields.append(DataField(c + 1, f"v{c}", AtomicType("INT")))
    schema = TableSchema(
        id=0, fields=fields, highest_field_id=len(fields),
        partition_keys=[], primary_keys=["k"], options={})

    mf = None
    if merge_function_name == "deduplicate":
        from pypaimon.read.reader.deduplicate_merge_function import \
            DeduplicateMergeFunction
        mf = DeduplicateMergeFunction()
    elif merge_function_name == "partial-update":
        from pypaimon.read.reader.partial_update_merge_function import \
            PartialUpdateMergeFunction
        mf = PartialUpdateMergeFunction(key_arity, n_value_cols)
    elif merge_function_name == "first-row":
        from pypaimon.read.reader.first_row_merge_function import \
            FirstRowMergeFunction
        mf = FirstRowMergeFunction(ignore_delete=False)
    else:
        raise ValueError(merge_function_name)

    reader = SortMergeReaderWithMinHeap(
        [MemReader(r) for r in runs], schema, merge_function=mf)
    out = []
    while True:
        it = reader.read_batch()
        if it is None:
            break
        while True:
            kv = it.next()
            if kv is None:
                break
            if drop_delete and not kv.is_add():
                continue
            key = kv.key.get_field(0)
            vals = tuple(kv.value.get_field(c) for c in range(n_value_cols))
            out.append((key, kv.sequence_number, kv.value_row_kind_byte, vals))
    reader.close()
    return out
