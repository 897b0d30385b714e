"""Bond perception from 3D coordinates — the role the reference fills
with its vendored third-party xyz2mol.py (reference
utils/descriptors_and_embeddings/xyz2mol.py:869 xyz2mol): turn
(atomic numbers, positions) into a molecular GRAPH with bond orders
and formal-charge estimates, without requiring rdkit.

This is an original implementation, not a port: bonds come from
covalent-radius distance criteria, bond orders from an iterative
valence-saturation pass (shortest unsaturated bonds promoted first),
formal charges from the remaining valence mismatch.  rdkit (absent in
this image) is only needed if you want an RDKit Mol / canonical
SMILES out of the perceived graph (``to_rdkit_mol``)."""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from ...data import Data

# Covalent radii in Angstrom (single-bond, standard published values,
# Cordero et al.-style) for elements the molecular datasets use.
COVALENT_RADII: Dict[int, float] = {
    1: 0.31, 2: 0.28, 3: 1.28, 4: 0.96, 5: 0.84, 6: 0.76, 7: 0.71,
    8: 0.66, 9: 0.57, 10: 0.58, 11: 1.66, 12: 1.41, 13: 1.21,
    14: 1.11, 15: 1.07, 16: 1.05, 17: 1.02, 18: 1.06, 19: 2.03,
    20: 1.76, 26: 1.32, 29: 1.32, 30: 1.22, 35: 1.20, 53: 1.39,
}

# Typical maximum valences.
MAX_VALENCE: Dict[int, int] = {
    1: 1, 3: 1, 5: 3, 6: 4, 7: 3, 8: 2, 9: 1, 11: 1, 12: 2, 13: 3,
    14: 4, 15: 5, 16: 6, 17: 1, 19: 1, 20: 2, 26: 6, 29: 4, 30: 2,
    35: 1, 53: 1,
}


def perceive_bonds(z: torch.Tensor, pos: torch.Tensor,
                   tolerance: float = 1.2
                   ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Distance-criterion bonds: (i, j) bonded when
    d_ij < tolerance * (r_i + r_j).  Returns (bonds [B, 2] with i<j,
    lengths [B])."""
    z = z.reshape(-1).long()
    pos = pos.reshape(-1, 3).double()
    n = z.numel()
    if n < 2:
        return (torch.zeros(0, 2, dtype=torch.long),
                torch.zeros(0, dtype=torch.float64))
    radii = torch.tensor([COVALENT_RADII.get(int(a), 1.5) for a in z],
                         dtype=torch.float64)
    d = torch.cdist(pos, pos)
    cut = tolerance * (radii.view(-1, 1) + radii.view(1, -1))
    mask = (d < cut) & (d > 1e-6)
    mask = torch.triu(mask, diagonal=1)
    bonds = mask.nonzero()
    return bonds, d[bonds[:, 0], bonds[:, 1]]


def assign_bond_orders(z: torch.Tensor, bonds: torch.Tensor,
                       lengths: torch.Tensor
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Iterative valence saturation: start all single; promote the
    shortest bond whose BOTH endpoints still have free valence, until
    no promotion is possible (max order 3).  Returns (orders [B],
    formal_charge_estimate [N] = valence deficit sign)."""
    z = z.reshape(-1).long()
    n = z.numel()
    nb = bonds.shape[0]
    orders = torch.ones(nb, dtype=torch.long)
    maxv = torch.tensor([MAX_VALENCE.get(int(a), 4) for a in z])
    used = torch.zeros(n, dtype=torch.long)
    for b in range(nb):
        used[bonds[b, 0]] += 1
        used[bonds[b, 1]] += 1
    # promotion loop, shortest bonds first (stronger = shorter)
    order_idx = torch.argsort(lengths)
    changed = True
    while changed:
        changed = False
        for b in order_idx.tolist():
            if orders[b] >= 3:
                continue
            i, j = int(bonds[b, 0]), int(bonds[b, 1])
            if used[i] < maxv[i] and used[j] < maxv[j]:
                orders[b] += 1
                used[i] += 1
                used[j] += 1
                changed = True
    charge_est = (used - maxv).clamp(min=-1, max=1)
    return orders, charge_est


def xyz_to_graph(z: torch.Tensor, pos: torch.Tensor,
                 tolerance: float = 1.2,
                 y: Optional[torch.Tensor] = None) -> Data:
    """Full perception pipeline: Data with bidirectional
    ``edge_index``, ``edge_attr`` = [bond_order, length], node features
    x = [Z, estimated formal charge]."""
    bonds, lengths = perceive_bonds(z, pos, tolerance)
    orders, charges = assign_bond_orders(z, bonds, lengths)
    src = torch.cat([bonds[:, 0], bonds[:, 1]])
    dst = torch.cat([bonds[:, 1], bonds[:, 0]])
    ei = torch.stack([src, dst], dim=0)
    ea = torch.cat([
        torch.stack([orders.float(), lengths.float()], dim=1),
        torch.stack([orders.float(), lengths.float()], dim=1)])
    # dst-major sort for the CSR fast path
    perm = torch.argsort(dst * (z.numel() + 1) + src)
    d = Data(
        x=torch.stack([z.float().reshape(-1),
                       charges.float()], dim=1),
        z=z.reshape(-1).long(),
        pos=pos.reshape(-1, 3).float(),
        edge_index=ei[:, perm].contiguous(),
        edge_attr=ea[perm].contiguous(),
    )
    if y is not None:
        d.y = y
    d.num_nodes = z.numel()
    return d


def to_rdkit_mol(z: torch.Tensor, pos: torch.Tensor,
                 tolerance: float = 1.2):
    """Perceived graph -> RDKit RWMol (requires rdkit; gives canonical
    SMILES via Chem.MolToSmiles).  The reference reaches the same end
    through its vendored xyz2mol."""
    try:
        from rdkit import Chem
        from rdkit.Geometry import Point3D
    except ImportError as e:  # pragma: no cover - rdkit absent here
        raise ImportError("to_rdkit_mol requires rdkit") from e
    bonds, lengths = perceive_bonds(z, pos, tolerance)
    orders, _ = assign_bond_orders(z, bonds, lengths)
    mol = Chem.RWMol()
    for a in z.reshape(-1).tolist():
        mol.AddAtom(Chem.Atom(int(a)))
    btype = {1: Chem.BondType.SINGLE, 2: Chem.BondType.DOUBLE,
             3: Chem.BondType.TRIPLE}
    for (i, j), o in zip(bonds.tolist(), orders.tolist()):
        mol.AddBond(int(i), int(j), btype[int(o)])
    conf = Chem.Conformer(mol.GetNumAtoms())
    for idx, p in enumerate(pos.reshape(-1, 3).tolist()):
        conf.SetAtomPosition(idx, Point3D(*p))
    mol.AddConformer(conf)
    return mol
