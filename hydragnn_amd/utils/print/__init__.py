from .print_utils import (
    print_master,
    print_distributed,
    iterate_tqdm,
    setup_log,
    log,
    log0,
)
