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
from .graph_samples_checks_and_updates import (  # noqa: F401,E402
    RadiusGraph, RadiusGraphPBC, RadiusInteractionGraphCPU, PBCDistance,
    PBCLocalCartesian, get_radius_graph, get_radius_graph_pbc,
    get_radius_graph_config, get_radius_graph_pbc_config, gather_deg,
    calculate_avg_deg, check_data_samples_equivalence,
    update_predicted_values, update_atom_features,
    should_skip_self_loops)
from .dataset_descriptors import AtomFeatures, StructureFeatures  # noqa: F401,E402
from .load_data import (  # noqa: F401,E402
    SimpleDataLoader, load_train_val_test_sets,
    total_to_train_val_test_pkls)
from .compositional_splitting import (  # noqa: F401,E402
    compositional_stratified_splitting)
