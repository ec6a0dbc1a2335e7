from .gpu_replay import ReplayShard, ShardedReplay  # noqa: F401
