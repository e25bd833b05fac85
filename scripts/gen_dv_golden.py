#!/usr/bin/env python3
"""Generate deletion-vector fixtures under tests/golden/.

The bitmap bytes follow the PUBLISHED portable Roaring serialization
(RoaringFormatSpec: cookie 12347 = no run containers, cookie low-16 12346 =
with runs; little-endian; array/bitmap/run containers), restated here
independently of the C parser in plan.cpp. The wrapper is the reference's
BitmapDeletionVector on-disk form (BitmapDeletionVector.java:98-112:
[i32 BE size][i32 BE magic 1581511376][bitmap][i32 BE crc32]). pypaimon's
own serializer delegates to the absent pyroaring package, so this
restatement is the pin (stated in DESIGN.md §5); the GPU tests additionally
verify END-TO-END row filtering against numpy-filtered runs, which is
format-independent."""
import json
import os
import struct
import sys
import zlib

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def serialize_roaring32(values, use_runs=False):
    """Portable Roaring serialization of a sorted set of u32."""
    values = sorted(set(int(v) for v in values))
    conts = {}
    for v in values:
        conts.setdefault(v >> 16, []).append(v & 0xFFFF)
    keys = sorted(conts)
    n = len(keys)
    bodies = []
    run_flags = bytearray((n + 7) // 8)
    for i, k in enumerate(keys):
        vals = conts[k]
        # run-length encode to decide representation
        runs = []
        s = p = vals[0]
        for v in vals[1:]:
            if v == p + 1:
                p = v
            else:
                runs.append((s, p - s))
                s = p = v
        runs.append((s, p - s))
        if use_runs and len(runs) * 4 + 2 < min(8192, 2 * len(vals)):
            body = struct.pack("<H", len(runs))
            for s0, ln in runs:
                body += struct.pack("<HH", s0, ln)
            run_flags[i // 8] |= 1 << (i % 8)
        elif len(vals) > 4096:
            bits = bytearray(8192)
            for v in vals:
                bits[v // 8] |= 1 << (v % 8)
            body = bytes(bits)
        else:
            body = b"".join(struct.pack("<H", v) for v in vals)
        bodies.append(body)
    out = b""
    has_run = use_runs and any(run_flags)
    if has_run:
        out += struct.pack("<I", (12346 | ((n - 1) << 16)) & 0xFFFFFFFF)
        out += bytes(run_flags)
    else:
        out += struct.pack("<I", 12347)
        out += struct.pack("<I", n)
    for i, k in enumerate(keys):
        out += struct.pack("<HH", k, len(conts[k]) - 1)
    if not has_run or n >= 4:
        off = len(out) + 4 * n
        for b in bodies:
            out += struct.pack("<I", off)
            off += len(b)
    for b in bodies:
        out += b
    return out


def wrap_dv(bitmap_bytes):
    """BitmapDeletionVector.serializeTo: [size BE][magic BE + bitmap][crc BE]."""
    data = struct.pack(">i", 1581511376) + bitmap_bytes
    return struct.pack(">i", len(data)) + data + \
        struct.pack(">i", zlib.crc32(data) & 0x7FFFFFFF)


def main():
    rng = np.random.default_rng(77)
    cases = {
        "small_scattered": (5000, sorted(
            rng.choice(5000, 700, replace=False).tolist()), False),
        "dense_block": (20000, list(range(3000, 15000)), False),
        "sparse": (20000, sorted(
            rng.choice(20000, 60, replace=False).tolist()), False),
        "first_last": (1000, [0, 999], False),
        "run_form": (30000, list(range(100, 9000)) + [25000, 29999], True),
        "wide": (200000, sorted(
            rng.choice(200000, 5000, replace=False).tolist()), False),
    }
    blob = b""
    index = {}
    for name, (rows, pos, runs) in cases.items():
        ser = wrap_dv(serialize_roaring32(pos, use_runs=runs))
        index[name] = {"offset": len(blob), "length": len(ser),
                       "rows": rows, "cardinality": len(set(pos))}
        blob += ser
        np.save(os.path.join(REPO, "tests", "golden", f"dv_pos_{name}.npy"),
                np.array(sorted(set(pos)), dtype=np.int64))
    with open(os.path.join(REPO, "tests", "golden", "dv_index.bin"),
              "wb") as f:
        f.write(blob)
    with open(os.path.join(REPO, "tests", "golden", "dv_index.json"),
              "w") as f:
        json.dump(index, f, indent=1)
    print(json.dumps({k: v["cardinality"] for k, v in index.items()}))


if __name__ == "__main__":
    main()
