from novel_view_synthesis_3d_amd.parallel.ddp import (  # noqa: F401
    DataParallelEngine, init_distributed, distributed_info,
)
