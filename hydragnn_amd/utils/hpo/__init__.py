from .deephyper import (
    parse_slurm_nodelist,
    master_from_host,
    run_random_search,
    read_node_list,
)
