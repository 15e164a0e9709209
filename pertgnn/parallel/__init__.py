from .comm import Comm
from .ddp import GradBucketAllReduce

__all__ = ["Comm", "GradBucketAllReduce"]
