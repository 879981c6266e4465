from shifu_amd.parallel.dist import (  # noqa: F401
    init_distributed, destroy_distributed, GradAggregator, is_distributed,
)
