"""Feature-column enums (reference
hydragnn/preprocess/dataset_descriptors.py): names paired with column
indexes used by raw-dataset loaders and update_atom_features."""

from enum import Enum


class AtomFeatures(Enum):
    NUM_OF_PROTONS = 0
    CHARGE_DENSITY = 1
    MAGNETIC_MOMENT = 2


class StructureFeatures(Enum):
    FREE_ENERGY = 0
    CHARGE_DENSITY = 1
    MAGNETIC_MOMENT = 2
