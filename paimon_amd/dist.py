"""Bucket sharding across GPUs — SURVEY.md §8e: one bucket = one independent
merge problem (DataSplit unit, table/source/DataSplit.java:63-75); buckets
round-robin onto ranks; no data-path collective. torch.distributed (RCCL on
GPU, gloo in CPU tests) is used only for the timing barrier and the
max-over-ranks reduction."""


def shard_buckets(n_buckets: int, world_size: int, rank: int):
    """Round-robin bucket ids for this rank (deterministic, collective-free)."""
    return list(range(rank, n_buckets, world_size))


def aggregate_rows_per_sec(rows_per_rank, elapsed_max_s):
    """Whole-job throughput: all ranks' rows over the max-of-ranks elapsed."""
    return sum(rows_per_rank) / elapsed_max_s
