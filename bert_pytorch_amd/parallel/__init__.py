from .comm import (  # noqa: F401
    WorkerInitObj,
    barrier,
    get_rank,
    get_world_size,
    init_distributed,
    is_main_process,
    mkdir_by_main_process,
    wrap_ddp,
)
