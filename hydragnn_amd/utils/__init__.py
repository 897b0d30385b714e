from . import config, datasets, distributed, model, optimizer, print
from . import profiling_and_tracing

# convenience re-exports at hydragnn_amd.utils.<name> (older reference
# scripts address helpers this way, e.g. utils.setup_ddp())
from .config import update_config, get_log_name_config, save_config  # noqa: F401
from .distributed import (  # noqa: F401
    setup_ddp, get_comm_size_and_rank, get_device,
    distributed_model_wrapper, get_distributed_model)
from .model import (  # noqa: F401
    save_model, load_existing_model, load_existing_model_config,
    get_summary_writer)
from .print.print_utils import setup_log, print_distributed, iterate_tqdm  # noqa: F401
