from .lsms import (
    convert_raw_data_energy_to_gibbs,
    get_formation_enthalpy,
    compositional_histogram_cutoff,
)
