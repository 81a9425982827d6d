from .dist import init_from_env, get_rank_world, barrier  # noqa: F401
from .shard import exchange_batch_all_to_all, shard_scan  # noqa: F401
