from .preprocessing import (
    voigt_to_full,
    normalize_stress,
    validate_atomistic_sample,
)
