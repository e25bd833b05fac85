"""Shared test helpers: fixture loading and the textual KV-stream parser
ported from the reference's ReusingTestData.parse
(paimon-core/src/test/java/org/apache/paimon/utils/ReusingTestData.java:90-106):
streams split on '|', each record 'key, seq, + or -, value'."""

import numpy as np

KIND_INSERT = 0
KIND_DELETE = 3


def parse_stream(s):
    ks, ss, kd, vv = [], [], [], []
    for kv in s.split("|"):
        kv = kv.strip()
        if not kv:
            continue
        p = [x.strip() for x in kv.split(",")]
        assert len(p) == 4, f"invalid data string {kv}"
        ks.append(int(p[0]))
        ss.append(int(p[1]))
        kd.append(KIND_INSERT if p[2] == "+" else KIND_DELETE)
        vv.append(int(p[3]))
    return {
        "key": np.array(ks, np.int64),
        "seq": np.array(ss, np.int64),
        "kind": np.array(kd, np.int8),
        "values": [np.array(ks, np.int64), np.array(vv, np.int64)],
    }


def parse_runs(*streams):
    return [parse_stream(s) for s in streams]


def load_runs_npz(d, prefix="in"):
    n = int(d[f"{prefix}_n_runs"])
    runs = []
    for i in range(n):
        r = {"key": d[f"{prefix}_key_{i}"], "seq": d[f"{prefix}_seq_{i}"],
             "kind": d[f"{prefix}_kind_{i}"]}
        nv_key = f"{prefix}_nvals_{i}"
        if nv_key in d:
            nv = int(d[nv_key])
            r["values"] = [d[f"{prefix}_val_{i}_{c}"] for c in range(nv)]
            if f"{prefix}_msk_{i}_0" in d:
                r["valid"] = [d[f"{prefix}_msk_{i}_{c}"] for c in range(nv)]
        runs.append(r)
    return runs


def random_runs(rng, n_runs, max_rows, key_space, delete_p=0.2, n_value_cols=0,
                null_p=0.0):
    """Random sorted runs obeying LSM invariants (sorted, unique keys per
    run, globally unique seqs). null_p > 0 adds 'valid' masks (col 0, the
    key copy, stays non-null)."""
    total_cap = n_runs * max_rows
    seqpool = rng.permutation(total_cap * 2).astype(np.int64)
    off = 0
    runs = []
    for _ in range(n_runs):
        n = int(rng.integers(0, max_rows + 1))
        keys = np.sort(rng.choice(key_space, size=min(n, key_space),
                                  replace=False)).astype(np.int64)
        n = len(keys)
        r = {"key": keys, "seq": seqpool[off:off + n],
             "kind": rng.choice([0, 3], n, p=[1 - delete_p, delete_p]).astype(np.int8)}
        if n_value_cols:
            r["values"] = [keys.copy()] + [
                rng.integers(-2**31, 2**31, n).astype(np.int32)
                for _ in range(n_value_cols)]
            if null_p > 0:
                r["valid"] = [np.ones(n, dtype=bool)] + [
                    rng.random(n) >= null_p for _ in range(n_value_cols)]
        off += n
        runs.append(r)
    return runs
