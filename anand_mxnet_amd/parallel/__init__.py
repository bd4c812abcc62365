"""Parallelism utilities: bucketed gradient all-reduce over RCCL (xGMI)."""
from .bucketed import BucketedAllReducer, GradBucket, DEFAULT_BUCKET_BYTES
