"""ORACLE package — CPU restatement of Paimon's merge-on-read hot path.

TEST INFRASTRUCTURE ONLY: only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this package. The product path (paimon_amd +
libpaimon_hip.so) must never route through it.
"""
from .oracle import (  # noqa: F401
    merge_order,
    merge_dedup,
    merge_dedup_count_mt,
    merge_dedup_model,
    merge_dedup_useq_model, full_changelog_model,
    merge_first_row_model,
    partial_update_model,
    partial_update_rrod_model,
    partial_update_seqgroup_model,
    aggregation_model,
    aggregation_rrod_model,
    aggregation_retract_model,
    rle_bp_decode,
    lib_path,
)
