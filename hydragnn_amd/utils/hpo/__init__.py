from .deephyper import (
    parse_slurm_nodelist,
    master_from_host,
    run_random_search,
    run_deephyper_search,
    run_search,
    read_node_list,
)
