"""paimon_amd — MI355X-native implementation of Apache Paimon's merge-on-read
hot path: LSM sorted-run k-way merge (Deduplicate/PartialUpdate) fed by
Parquet column-chunk decode, as hand-written CDNA4 HIP kernels behind a
C-ABI (libpaimon_hip.so). See DESIGN.md and include/paimon_hip.h."""

from .reader import (  # noqa: F401
    Session,
    MergeReadPlan,
    load_lib,
    debug_footer,
    interval_partition,
    file_descs_from_metas,
    write_parquet,
    LIB_PATH,
)

__version__ = "0.1.0"
