"""Name-compatibility aliases for reference flows that import
AdiosWriter / AdiosDataset (reference adiosdataset.py:58,365).

There is no ADIOS2 on this stack; the MI355X-native GraphStore
(graphstore.py) provides the same roles — per-key global arrays with
variable-dim count/offset metadata, attribute store, preload / memmap
read modes, and the DDStore-style DistDataset fetch window."""

from .graphstore import DistDataset  # noqa: F401
from .graphstore import GraphStoreDataset as AdiosDataset  # noqa: F401
from .graphstore import GraphStoreWriter as AdiosWriter  # noqa: F401
