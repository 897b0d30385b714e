"""SMILES -> graph conversion (reference: hydragnn/utils/
descriptors_and_embeddings/smiles_utils.py:28-137 + vendored
xyz2mol.py).  Requires rdkit, which is not in this image; the functions
raise a clear ImportError so config-driven flows fail loudly instead of
silently."""

from __future__ import annotations

from typing import List, Tuple

import torch

from ...data import Data


def _require_rdkit():
    try:
        from rdkit import Chem  # noqa: F401
        return Chem
    except ImportError as e:  # pragma: no cover
        raise ImportError(
            "SMILES utilities require rdkit, which is not installed in "
            "this environment. Install rdkit to use "
            "generate_graphdata_from_smilestr.") from e


def get_node_attribute_name(types=None) -> Tuple[List[str], List[int]]:
    types = types or ["C", "N", "O", "F", "H"]
    names = [f"atom_type_{t}" for t in types] + ["degree", "charge",
                                                 "aromatic"]
    dims = [1] * len(names)
    return names, dims


def generate_graphdata_from_smilestr(smilestr: str, ytarget,
                                     types=None) -> Data:
    Chem = _require_rdkit()
    mol = Chem.MolFromSmiles(smilestr)
    if mol is None:
        raise ValueError(f"invalid SMILES: {smilestr}")
    return generate_graphdata_from_rdkit_molecule(mol, ytarget, types)


def generate_graphdata_from_rdkit_molecule(mol, ytarget,
                                           types=None) -> Data:
    """Graph Data from an rdkit molecule (reference
    smiles_utils.py:59)."""
    Chem = _require_rdkit()
    types = types or ["C", "N", "O", "F", "H"]
    mol = Chem.AddHs(mol)
    n = mol.GetNumAtoms()
    feats = []
    for atom in mol.GetAtoms():
        onehot = [1.0 if atom.GetSymbol() == t else 0.0 for t in types]
        feats.append(onehot + [atom.GetDegree() / 4.0,
                               float(atom.GetFormalCharge()),
                               float(atom.GetIsAromatic())])
    src, dst = [], []
    for bond in mol.GetBonds():
        i, j = bond.GetBeginAtomIdx(), bond.GetEndAtomIdx()
        src += [i, j]
        dst += [j, i]
    data = Data(
        x=torch.tensor(feats),
        edge_index=torch.tensor([src, dst], dtype=torch.long),
        y=torch.as_tensor(ytarget).view(-1, 1).float(),
    )
    data.num_nodes = n
    return data
