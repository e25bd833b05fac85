#!/usr/bin/env python3
"""Write-side zstd A/B on the GPU box: write_parquet(compression=zstd)
with the GPU block compressor vs the host libzstd codec."""
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
from paimon_amd.reader import write_parquet  # noqa: E402

n = 20_000_000
k = np.arange(n, dtype=np.int64) * 3
cols = [("_KEY_k", k), ("_SEQUENCE_NUMBER", k.copy()),
        ("_VALUE_KIND", np.zeros(n, np.int8)),
        ("v_k", (k * 7) % 1000003)] + \
       [(f"v_c{i}", ((k // (i + 2)) % 97).astype(np.int32))
        for i in range(4)]
for tag in ("gpu", "host"):
    os.environ["PMH_GPU_ZSTD_ENC"] = "1" if tag == "gpu" else "0"
    # env read once per process; fork a child per leg
    pid = os.fork()
    if pid == 0:
        t0 = time.perf_counter()
        write_parquet(f"/tmp/ab_{tag}.parquet", cols, compression="zstd")
        dt = time.perf_counter() - t0
        sz = os.path.getsize(f"/tmp/ab_{tag}.parquet")
        print(f"{tag}: {dt:.2f}s  file {sz/1e6:.1f} MB", flush=True)
        os._exit(0)
    os.waitpid(pid, 0)
# cross-check: both files must read back identically with pyarrow
import pyarrow.parquet as pq  # noqa: E402
a = pq.read_table("/tmp/ab_gpu.parquet")
b = pq.read_table("/tmp/ab_host.parquet")
assert a.equals(b), "content mismatch"
print("pyarrow reads both, contents equal")
