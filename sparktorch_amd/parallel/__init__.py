from sparktorch_amd.parallel.buckets import FlatBuckets
from sparktorch_amd.parallel.sync import SyncTrainer, train_distributed
