from .base import Base, MLPNode
from .create import create_model, create_model_config, EnhancedModelWrapper
from .stacks import (
    CGCNNStack, GATStack, GINStack, MFCStack, PNAStack, SAGEStack,
)
