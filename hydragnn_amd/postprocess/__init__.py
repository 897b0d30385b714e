from .postprocess import output_denormalize, unscale_features_by_num_nodes
from .visualizer import Visualizer
