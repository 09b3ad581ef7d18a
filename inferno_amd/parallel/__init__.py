from .dist import ShardedSolver, ShardResult, shard_servers  # noqa: F401
