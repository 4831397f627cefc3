from .comm import Comm
from .ddp import BucketedDDP, GradBucket, partition_buckets
from .fake_pg import RecordingComm

__all__ = ["Comm", "BucketedDDP", "GradBucket", "partition_buckets", "RecordingComm"]
