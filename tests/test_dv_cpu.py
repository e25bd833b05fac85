"""Deletion-vector parser (plan.cpp parse_roaring32 + the
BitmapDeletionVector.java:98-112 wrapper) pinned against independently
serialized fixtures under tests/golden/ (scripts/gen_dv_golden.py restates
the published portable Roaring format: array, bitmap and run containers,
both cookie forms)."""

import json
import os

import numpy as np

from paimon_amd.reader import debug_parse_dv, load_lib
import pytest

GOLD = os.path.join(os.path.dirname(__file__), "golden")


@pytest.mark.skipif(load_lib(required=False) is None,
                    reason="libpaimon_hip.so not built")
class TestDvParser:
    def test_all_fixture_forms(self):
        idx = json.load(open(os.path.join(GOLD, "dv_index.json")))
        assert len(idx) >= 5
        for name, e in idx.items():
            got = debug_parse_dv(os.path.join(GOLD, "dv_index.bin"),
                                 e["offset"], e["length"])
            exp = np.load(os.path.join(GOLD, f"dv_pos_{name}.npy"))
            assert len(got) == e["cardinality"], name
            assert (got == exp).all(), name

    def test_bad_magic(self, tmp_path):
        p = tmp_path / "bad.bin"
        p.write_bytes(b"\x00\x00\x00\x08" + b"\x00" * 12)
        with pytest.raises(RuntimeError, match="magic"):
            debug_parse_dv(str(p), 0, 16)
