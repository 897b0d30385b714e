from .base import Base, MLPNode
from .create import create_model, create_model_config, EnhancedModelWrapper
from .stacks import (
    CGCNNStack, GATStack, GINStack, MFCStack, PNAStack, SAGEStack,
)
from .pna_plus import PNAPlusStack
from .schnet import SCFStack
from .dimenet import DIMEStack
from .egnn import EGCLStack
from .painn import PAINNStack
from .pnaeq import PNAEqStack
from .mace import MACEStack
from .multitask_mp import MultiTaskModelMP, DualOptimizer
