"""Sequence-group model pinned against the reference's own executable test
vectors (PartialUpdateMergeFunctionTest.testSequenceGroup, ported literally)
— fields f3 -> group {f1,f2}, f6 -> group {f4,f5}, value row f0..f6."""

import numpy as np

from oracle import partial_update_seqgroup_model

SGS = [{"sequence_fields": [3], "group_fields": [1, 2]},
       {"sequence_fields": [6], "group_fields": [4, 5]}]


def _runs(records):
    """records: list of (kind, f0..f6 with None) — one 1-row run each, same
    key, ascending seq (the reference test feeds one func sequentially)."""
    runs = []
    for s, rec in enumerate(records):
        kind = rec[0]
        vals = rec[1:]
        r = {"key": np.array([1], np.int64),
             "seq": np.array([s], np.int64),
             "kind": np.array([kind], np.int8),
             "values": [np.array([0 if v is None else v], np.int32)
                        for v in vals],
             "valid": [np.array([v is not None]) for v in vals]}
        runs.append(r)
    return runs


def _row(out):
    assert len(out["key"]) == 1
    return tuple(int(out["values"][c][0]) if out["valid"][c][0] else None
                 for c in range(7))


RECORDS = [
    (0, 1, 1, 1, 1, 1, 1, 1),
    (0, 1, 2, 2, 2, 2, 2, None),
    (0, 1, 3, 3, 1, 3, 3, 3),
    (3, 1, 1, 1, 3, 1, 1, None),   # DELETE
    (3, 1, 1, 1, 3, 1, 1, 4),      # DELETE
    (0, 1, 4, 4, 4, 5, 5, 5),
    (3, 1, 1, 1, 6, 1, 1, 6),      # DELETE
]

# validate(...) states after each prefix, PartialUpdateMergeFunctionTest
# testSequenceGroup (prefix of length 1 is the wrapper bypass, not checked
# by the reference test; start at 2)
EXPECTED = {
    2: (1, 2, 2, 2, 1, 1, 1),
    3: (1, 2, 2, 2, 3, 3, 3),
    4: (1, None, None, 3, 3, 3, 3),
    5: (1, None, None, 3, None, None, 4),
    6: (1, 4, 4, 4, 5, 5, 5),
    7: (1, None, None, 6, None, None, 6),
}


class TestReferenceVectors:
    def test_sequence_group_prefixes(self):
        for n, exp in EXPECTED.items():
            out = partial_update_seqgroup_model(_runs(RECORDS[:n]), SGS,
                                                drop_delete=False)
            assert _row(out) == exp, n

    def test_kind_delete_when_no_insert(self):
        # a group of only retracts folds to RowKind.DELETE (getResult)
        recs = [(3, 1, 1, 1, 3, 1, 1, None), (3, 1, 1, 1, 4, 1, 1, None)]
        out = partial_update_seqgroup_model(_runs(recs), SGS,
                                            drop_delete=False)
        assert out["kind"][0] == 3
        out = partial_update_seqgroup_model(_runs(recs), SGS,
                                            drop_delete=True)
        assert len(out["key"]) == 0

    def test_multi_field_sequence(self):
        # two sequence fields compare lexicographically, nulls first
        sgs = [{"sequence_fields": [2, 3], "group_fields": [1]}]
        recs = [(0, 1, 10, 2, 1, 0, 0, 0),
                (0, 1, 20, 2, 0, 0, 0, 0),   # (2,0) < (2,1): ignored
                (0, 1, 30, None, 5, 0, 0, 0)]  # (null,5) < (2,1): ignored
        out = partial_update_seqgroup_model(_runs(recs), sgs,
                                            drop_delete=False)
        assert _row(out)[1] == 10
        recs.append((0, 1, 40, 2, 1, 0, 0, 0))  # tie: accepted (>=)
        out = partial_update_seqgroup_model(_runs(recs), sgs,
                                            drop_delete=False)
        assert _row(out)[1] == 40
