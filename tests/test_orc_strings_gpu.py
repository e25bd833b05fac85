"""ORC string columns on the GPU path: DIRECT_V2 (LENGTH + DATA bytes,
interned into the plan-level global dictionary at staging) and
DICTIONARY_V2 (stripe dictionary remapped to the global dictionary, RLEv2
ids decoded on device and remapped in place) — StringTreeWriter's two
encodings. Parity against the oracle over the original string arrays."""
import os

import numpy as np
import pyarrow as pa
import pyarrow.orc as orc
import pytest

from oracle import merge_dedup
from paimon_amd import Session, MergeReadPlan, file_descs_from_metas
from paimon_amd.datagen import gen_runs_dedup

pytestmark = pytest.mark.gpu

KEY_COLS = [{"name": "_KEY_k", "type": "int64"}]


def _write_orc_str(runs, strs, out_dir, dict_threshold, null_masks=None,
                   compression="uncompressed"):
    os.makedirs(out_dir, exist_ok=True)
    metas = []
    for i, r in enumerate(runs):
        cols = {
            "_KEY_k": pa.array(r["key"]),
            "_SEQUENCE_NUMBER": pa.array(r["seq"]),
            "_VALUE_KIND": pa.array(r["kind"].astype(np.int8)),
            "v_k": pa.array(r["values"][0]),
            "v_s": pa.array(strs[i],
                            mask=null_masks[i] if null_masks else None),
            "v_c0": pa.array(r["values"][1]),
        }
        path = os.path.join(out_dir, f"run-{i}.orc")
        orc.write_table(pa.table(cols), path, compression=compression,
                        dictionary_key_size_threshold=dict_threshold)
        metas.append({"path": path, "rowCount": len(r["key"]),
                      "minKey": int(r["key"][0]),
                      "maxKey": int(r["key"][-1]), "level": 0})
    return metas


VALUE_COLS = [{"name": "v_k", "type": "int64"},
              {"name": "v_s", "type": "string"},
              {"name": "v_c0", "type": "int32"}]


def _check(tmp_path, dict_threshold, card=40, null_frac=0.0, n_runs=4,
           rows=25_000, seed=801, compression="uncompressed"):
    rng = np.random.default_rng(seed)
    runs = gen_runs_dedup(n_runs, rows, n_value_cols=1, seed=seed,
                          delete_frac=0.15)
    vocab = np.array([f"str-{j:04d}-{'x' * (j % 9)}" for j in range(card)])
    strs, masks = [], []
    for r in runs:
        pick = rng.integers(0, card, len(r["key"]))
        strs.append(vocab[pick].tolist())
        masks.append(rng.random(len(r["key"])) < null_frac
                     if null_frac else None)
    metas = _write_orc_str(runs, strs, str(tmp_path), dict_threshold,
                           null_masks=masks if null_frac else None,
                           compression=compression)
    rr, ww = merge_dedup(runs)
    exp_keys = np.array([runs[a]["key"][b] for a, b in zip(rr, ww)],
                        np.int64)
    exp_str = [strs[a][b] for a, b in zip(rr, ww)]
    exp_null = (np.array([masks[a][b] for a, b in zip(rr, ww)], bool)
                if null_frac else np.zeros(len(rr), bool))
    with Session(0) as s:
        with MergeReadPlan(s, file_descs_from_metas(metas), KEY_COLS,
                           VALUE_COLS) as plan:
            got = {}
            while True:
                b = plan.read_next()
                if b is None:
                    break
                for kk, v in b.items():
                    got.setdefault(kk, []).append(
                        v if kk.endswith("#dict") else v.copy())
            dicts = got.pop("v_s#dict")[-1]
            got = {kk: np.concatenate(v) for kk, v in got.items()}
    assert (got["_KEY_k"] == exp_keys).all()
    ids = got["v_s"]
    if null_frac:
        valid = got["v_s#valid"]
        assert (valid == ~exp_null).all()
    live = ~exp_null
    dec = np.array([dicts[i].decode() for i in ids[live]])
    assert (dec == np.array(exp_str, dtype=object)[live].astype(str)).all()


class TestOrcStrings:
    def test_dictionary_v2(self, tmp_path):
        _check(tmp_path, dict_threshold=1.0, card=40, seed=801)

    def test_direct_v2(self, tmp_path):
        # high cardinality + threshold 0 -> DIRECT_V2 (host id-ification)
        _check(tmp_path, dict_threshold=0.0, card=5000, seed=802)

    def test_dictionary_with_nulls(self, tmp_path):
        _check(tmp_path, dict_threshold=1.0, card=25, null_frac=0.3,
               seed=803)

    def test_direct_with_nulls(self, tmp_path):
        _check(tmp_path, dict_threshold=0.0, card=3000, null_frac=0.2,
               seed=804)

    def test_zstd_compressed_streams(self, tmp_path):
        _check(tmp_path, dict_threshold=1.0, card=30, seed=805,
               compression="zstd")

    def test_global_dict_spans_runs_and_encodings(self, tmp_path):
        # one run dictionary-encoded, one direct, shared global dictionary
        rng = np.random.default_rng(806)
        runs = gen_runs_dedup(2, 10_000, n_value_cols=1, seed=806)
        vocab = np.array([f"mix-{j}" for j in range(12)])
        strs = [vocab[rng.integers(0, 12, len(r["key"]))].tolist()
                for r in runs]
        m0 = _write_orc_str(runs[:1], strs[:1], str(tmp_path) + "/a", 1.0)
        m1 = _write_orc_str(runs[1:], strs[1:], str(tmp_path) + "/b", 0.0)
        rr, ww = merge_dedup(runs)
        exp_str = [strs[a][b] for a, b in zip(rr, ww)]
        with Session(0) as s:
            with MergeReadPlan(s, file_descs_from_metas(m0 + m1), KEY_COLS,
                               VALUE_COLS) as plan:
                got = {}
                dicts = None
                while True:
                    b = plan.read_next()
                    if b is None:
                        break
                    dicts = b["v_s#dict"]
                    got.setdefault("v_s", []).append(b["v_s"].copy())
                ids = np.concatenate(got["v_s"])
        dec = [dicts[i].decode() for i in ids]
        assert dec == [str(x) for x in exp_str]
