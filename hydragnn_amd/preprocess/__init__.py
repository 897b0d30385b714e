from .load_data import (
    create_dataloaders,
    dataset_loading_and_splitting,
    split_dataset,
)
from .batch_sampler import (
    CostAwareBatchSampler,
    DistributedCostAwareBatchSampler,
    graph_node_costs,
)
from ..ops import radius_graph, radius_graph_pbc
from .transforms import add_laplacian_pe, add_edge_lengths, normalize_rotation
from .transforms import pbc_as_tensor, pbc_distance, pbc_local_cartesian
from .dataloader import HydraDataLoader, parse_omp_places
from .energy_linear_regression import (
    energy_linear_regression,
    shift_energies,
)
from .stratified_sampling import stratified_sampling
from ..utils.config.config_utils import (
    check_if_graph_size_variable,
    _gather_deg as gather_deg,
)
