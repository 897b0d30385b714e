"""Materials preprocessing utilities (reference: hydragnn/utils/
materials/preprocessing.py:24-118): stress Voigt->full conversion with
unit/sign normalization to eV/A^3 tensile-positive, atomistic sample
schema validation with field-specific errors."""

from __future__ import annotations

from typing import Optional

import torch

KBAR_TO_EV_PER_A3 = 1.0 / 1602.1766208  # 1 kbar = 0.1 GPa
GPA_TO_EV_PER_A3 = 1.0 / 160.21766208


def voigt_to_full(stress: torch.Tensor) -> torch.Tensor:
    """[6] Voigt (xx, yy, zz, yz, xz, xy) -> [3,3] symmetric tensor."""
    s = stress.flatten()
    assert s.numel() == 6, "Voigt stress must have 6 components"
    return torch.tensor([
        [s[0], s[5], s[4]],
        [s[5], s[1], s[3]],
        [s[4], s[3], s[2]],
    ], dtype=stress.dtype)


def normalize_stress(stress: torch.Tensor, units: str = "eV/A3",
                     compressive_positive: bool = False) -> torch.Tensor:
    """Normalize stress to eV/A^3 with tensile-positive sign."""
    if stress.numel() == 6:
        stress = voigt_to_full(stress)
    stress = stress.reshape(3, 3)
    if units.lower() in ("kbar",):
        stress = stress * KBAR_TO_EV_PER_A3
    elif units.lower() in ("gpa",):
        stress = stress * GPA_TO_EV_PER_A3
    elif units.lower() not in ("ev/a3", "ev/ang3"):
        raise ValueError(f"unknown stress units {units}")
    if compressive_positive:
        stress = -stress
    return stress


def validate_atomistic_sample(data, require_forces: bool = False,
                              require_cell: bool = False) -> None:
    """Schema validation with field-specific errors (for distributed
    preprocessors, where a stack trace points at a rank not a field)."""
    pos = data.get("pos")
    if pos is None:
        raise ValueError("atomistic sample missing 'pos'")
    if pos.dim() != 2 or pos.shape[1] != 3:
        raise ValueError(f"'pos' must be [N,3], got {list(pos.shape)}")
    n = pos.shape[0]
    z = data.get("z")
    if z is not None and z.numel() != n:
        raise ValueError(
            f"'z' length {z.numel()} != num atoms {n}")
    forces = data.get("forces")
    if require_forces and forces is None:
        raise ValueError("atomistic sample missing 'forces'")
    if forces is not None and tuple(forces.shape) != (n, 3):
        raise ValueError(
            f"'forces' must be [{n},3], got {list(forces.shape)}")
    energy = data.get("energy")
    if energy is not None and energy.numel() != 1:
        raise ValueError("'energy' must be a scalar per sample")
    cell = data.get("cell") or data.get("supercell_size")
    if require_cell and cell is None:
        raise ValueError("periodic sample missing 'cell'")
    if cell is not None and cell.numel() != 9:
        raise ValueError("'cell' must be 3x3")
    if not torch.isfinite(pos).all():
        raise ValueError("'pos' contains non-finite values")


# reference-named alias
validate_materials_sample = validate_atomistic_sample
