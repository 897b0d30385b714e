"""hydragnn_amd.ops — CDNA4-native tensor primitives.

Dispatch policy: CUDA (ROCm) tensors run the hand-written HIP kernels in
the in-tree extension `_hip_ops` (built for gfx950); CPU tensors run
differentiable pure-PyTorch reference implementations.  On a GPU box a
missing extension raises — no silent eager fallback.
"""

from .scatter import scatter, gather, segment_softmax, degree
from .geometry import (
    get_edge_vectors_and_lengths,
    radius_graph,
    radius_graph_pbc,
)
from .sph import spherical_harmonics, sh_dim
from .basis import (
    bessel_basis,
    gaussian_basis,
    chebyshev_basis,
    sinc_basis,
    polynomial_cutoff,
    cosine_cutoff,
)
from ._extension import has_extension, get_extension
from .irreps_linear import irreps_linear
from .varlen_attn import varlen_attention
from .mfma_linear import MFMALinear
from .splitk_linear import SplitKLinear

__all__ = [
    "scatter", "gather", "segment_softmax", "degree",
    "get_edge_vectors_and_lengths", "radius_graph", "radius_graph_pbc",
    "spherical_harmonics", "sh_dim",
    "bessel_basis", "gaussian_basis", "chebyshev_basis", "sinc_basis",
    "polynomial_cutoff", "cosine_cutoff",
    "has_extension", "get_extension",
    "irreps_linear", "varlen_attention", "MFMALinear", "SplitKLinear",
]
