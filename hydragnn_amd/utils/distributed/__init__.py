from .distributed import (
    setup_ddp,
    get_device,
    get_device_name,
    get_comm_size_and_rank,
    get_local_rank,
    init_comm_size_and_rank,
    distributed_model_wrapper,
    get_distributed_model,
    nsplit,
    comm_reduce,
    print_peak_memory,
    check_remaining_time,
)
