from .dp import DataParallelPredictor, shard_sizes  # noqa: F401
