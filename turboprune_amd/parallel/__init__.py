from turboprune_amd.parallel.ddp import (  # noqa: F401
    setup_distributed,
    cleanup_distributed,
    broadcast_object,
    broadcast_model_state,
    check_model_equality,
    wrap_ddp,
    is_rank0,
    world_info,
)
