from melgan_amd.parallel.dp import (  # noqa: F401
    BucketAllReduceDP,
    init_distributed,
)
