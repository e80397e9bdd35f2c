from .vocab import EOS_ID, EOS_TOKEN, Vocab, load_dict
from .dataset import (
    Bucket,
    BucketLoader,
    Sample,
    load_pairs,
    make_buckets,
    prepare_batch,
    shard_buckets,
)
from .synth import fixed_shape_batch, synth_samples, write_synth_dataset

__all__ = [
    "EOS_ID", "EOS_TOKEN", "Vocab", "load_dict",
    "Bucket", "BucketLoader", "Sample", "load_pairs", "make_buckets",
    "prepare_batch", "shard_buckets",
    "fixed_shape_batch", "synth_samples", "write_synth_dataset",
]
