from .atomicdescriptors import atomicdescriptors
from .smiles_utils import get_node_attribute_name, generate_graphdata_from_smilestr
from .xyz2graph import (assign_bond_orders, perceive_bonds,
                        to_rdkit_mol, xyz_to_graph)
