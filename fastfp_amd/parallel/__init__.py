from fastfp_amd.parallel.dist import (  # noqa: F401
    all_gather_concat,
    cleanup,
    init_distributed,
    shard_slice,
)
