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
