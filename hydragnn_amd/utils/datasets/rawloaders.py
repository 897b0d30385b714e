"""Format-specific raw dataset loaders: LSMS, extended-XYZ, CFG.

Behavioral parity with /root/reference/hydragnn/utils/datasets/
{lsmsdataset.py,xyzdataset.py,cfgdataset.py} (the XYZ path drops the
ase dependency — extended-XYZ is parsed directly)."""

from __future__ import annotations

import re
from typing import Optional

import numpy as np
import torch

from ...data import Data
from .abstractrawdataset import AbstractRawDataset

_SYMBOLS = [
    "X", "H", "He", "Li", "Be", "B", "C", "N", "O", "F", "Ne", "Na", "Mg",
    "Al", "Si", "P", "S", "Cl", "Ar", "K", "Ca", "Sc", "Ti", "V", "Cr",
    "Mn", "Fe", "Co", "Ni", "Cu", "Zn", "Ga", "Ge", "As", "Se", "Br",
    "Kr", "Rb", "Sr", "Y", "Zr", "Nb", "Mo", "Tc", "Ru", "Rh", "Pd",
    "Ag", "Cd", "In", "Sn", "Sb", "Te", "I", "Xe", "Cs", "Ba", "La",
    "Ce", "Pr", "Nd", "Pm", "Sm", "Eu", "Gd", "Tb", "Dy", "Ho", "Er",
    "Tm", "Yb", "Lu", "Hf", "Ta", "W", "Re", "Os", "Ir", "Pt", "Au",
    "Hg", "Tl", "Pb", "Bi", "Po", "At", "Rn",
]
SYMBOL_TO_Z = {s: i for i, s in enumerate(_SYMBOLS)}


class LSMSDataset(AbstractRawDataset):
    """LSMS text format: line 0 = graph features; lines 1.. = per-atom
    rows with positions in columns 2-4 and features at the configured
    column indices (reference lsmsdataset.py:16-106)."""

    def transform_input_to_data_object_base(self, filepath) -> Optional[Data]:
        if not filepath.endswith((".txt", ".dat")) and "." in \
                filepath.rsplit("/", 1)[-1]:
            pass
        with open(filepath, "r", encoding="utf-8") as f:
            lines = f.readlines()
        if not lines:
            return None
        graph_feat = lines[0].split()
        g_feature = []
        for item in range(len(self.graph_feature_dim)):
            for icomp in range(self.graph_feature_dim[item]):
                g_feature.append(
                    float(graph_feat[self.graph_feature_col[item] + icomp]))
        node_features = []
        positions = []
        for line in lines[1:]:
            cols = line.split()
            if len(cols) < 5:
                continue
            positions.append([float(cols[2]), float(cols[3]),
                              float(cols[4])])
            feat = []
            for item in range(len(self.node_feature_dim)):
                for icomp in range(self.node_feature_dim[item]):
                    feat.append(
                        float(cols[self.node_feature_col[item] + icomp]))
            node_features.append(feat)
        data = Data(
            y=torch.tensor(g_feature),
            pos=torch.tensor(positions, dtype=torch.float),
            x=torch.tensor(node_features, dtype=torch.float),
        )
        return self._charge_density_update(data)

    def _charge_density_update(self, data: Data) -> Data:
        """charge density -> charge transfer: subtract the proton count
        (reference __charge_density_update_for_LSMS)."""
        if data.x.shape[1] >= 2:
            num_of_protons = data.x[:, 0]
            charge_density = data.x[:, 1]
            data.x[:, 1] = charge_density - num_of_protons
        return data


def parse_extxyz(filepath: str):
    """Minimal extended-XYZ parser: returns (symbols, positions, cell,
    info dict) for the first frame."""
    with open(filepath, "r", encoding="utf-8") as f:
        lines = f.read().splitlines()
    n = int(lines[0].split()[0])
    comment = lines[1] if len(lines) > 1 else ""
    info = {}
    for m in re.finditer(r'(\w+)=("[^"]*"|\S+)', comment):
        key, val = m.group(1), m.group(2).strip('"')
        info[key] = val
    cell = None
    if "Lattice" in info:
        vals = [float(v) for v in info["Lattice"].split()]
        cell = torch.tensor(vals, dtype=torch.float).view(3, 3)
    symbols, positions = [], []
    for line in lines[2:2 + n]:
        cols = line.split()
        symbols.append(cols[0])
        positions.append([float(cols[1]), float(cols[2]), float(cols[3])])
    return symbols, torch.tensor(positions, dtype=torch.float), cell, info


class XYZDataset(AbstractRawDataset):
    """Extended-XYZ loader (reference xyzdataset.py, ase-free)."""

    def transform_input_to_data_object_base(self, filepath) -> Optional[Data]:
        if not filepath.endswith(".xyz"):
            return None
        symbols, pos, cell, info = parse_extxyz(filepath)
        z = torch.tensor([SYMBOL_TO_Z.get(s, 0) for s in symbols],
                         dtype=torch.float)
        y_vals = []
        for name in self.graph_feature_name:
            if name in info:
                y_vals.append(float(info[name]))
        if not y_vals and "energy" in info:
            y_vals = [float(info["energy"])]
        data = Data(
            x=z.view(-1, 1),
            z=z.long(),
            pos=pos,
            y=torch.tensor(y_vals if y_vals else [0.0]),
        )
        if cell is not None:
            data.supercell_size = cell
        return data


class CFGDataset(AbstractRawDataset):
    """MTP .cfg configuration format loader (reference cfgdataset.py):
    BEGIN_CFG blocks with Size / Supercell / AtomData / Energy."""

    def transform_input_to_data_object_base(self, filepath) -> Optional[Data]:
        if not filepath.endswith(".cfg"):
            return None
        with open(filepath, "r", encoding="utf-8") as f:
            lines = [ln.strip() for ln in f.read().splitlines()]
        i = 0
        size = None
        cell_rows = []
        atoms = []
        forces_present = False
        energy = 0.0
        while i < len(lines):
            ln = lines[i]
            if ln.startswith("Size"):
                size = int(lines[i + 1])
                i += 2
                continue
            if ln.startswith(("Supercell", "SuperCell")):
                for j in range(3):
                    cell_rows.append([float(v)
                                      for v in lines[i + 1 + j].split()])
                i += 4
                continue
            if ln.startswith("AtomData"):
                header = ln.split()[1:]
                forces_present = "fx" in header
                for j in range(size):
                    atoms.append([float(v)
                                  for v in lines[i + 1 + j].split()])
                i += 1 + size
                continue
            if ln.startswith("Energy"):
                energy = float(lines[i + 1])
                i += 2
                continue
            i += 1
        if not atoms:
            return None
        arr = np.array(atoms)
        # columns: id type x y z [fx fy fz]
        data = Data(
            x=torch.tensor(arr[:, 1:2], dtype=torch.float),
            z=torch.tensor(arr[:, 1], dtype=torch.long),
            pos=torch.tensor(arr[:, 2:5], dtype=torch.float),
            y=torch.tensor([energy]),
            energy=torch.tensor([[energy]]),
        )
        if forces_present and arr.shape[1] >= 8:
            data.forces = torch.tensor(arr[:, 5:8], dtype=torch.float)
        if cell_rows:
            data.supercell_size = torch.tensor(cell_rows,
                                               dtype=torch.float)
        return data


# reference-named aliases (preprocess/*_raw_dataset_loader.py)
LSMS_RawDataLoader = LSMSDataset
CFG_RawDataLoader = CFGDataset
XYZ_RawDataLoader = XYZDataset
AbstractRawDataLoader = AbstractRawDataset
