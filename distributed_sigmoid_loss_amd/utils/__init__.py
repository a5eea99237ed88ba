from .seed import set_seed, seeded_global_batch, rank_shard
from .dist import (
    setup_process_group,
    cleanup_process_group,
    auto_backend,
    init_from_env,
)

__all__ = [
    "set_seed",
    "seeded_global_batch",
    "rank_shard",
    "setup_process_group",
    "cleanup_process_group",
    "auto_backend",
    "init_from_env",
]
