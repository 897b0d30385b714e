"""Reader/converter for reference HydraGNN ADIOS2 ``.bp`` datasets.

The reference writes datasets as ADIOS2 bp stores (reference
hydragnn/utils/datasets/adiosdataset.py:120-287 AdiosWriter.save):
per label (trainset/valset/testset)

  attributes: ``{label}/keys`` (string list), ``{label}/ndata``;
  global attributes: ``minmax_graph_feature``, ``minmax_node_feature``,
  ``pna_deg``, ``dataset_name``;
  per key ``k``: a concatenated global array ``{label}/{k}`` plus
  ``{label}/{k}/variable_count|variable_offset|variable_dim``.

This module reads that layout (optional ``import adios2`` — absent in
the MI355X image but available on reference clusters) into
``hydragnn_amd.data.Data`` samples and converts whole stores into the
native GraphStore format so existing reference datasets keep working:

    python -m hydragnn_amd.utils.datasets.adios_reader \
        --bp dataset.bp --out ./graphstore_dir

The ADIOS2 API surface used is isolated in ``_Adios2File`` so the
logic is testable without adios2 (tests/test_adios_reader.py mocks
it).
"""

from __future__ import annotations

import argparse
import os
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from ...data import Data

LABELS = ("trainset", "valset", "testset")


class _Adios2File:
    """Thin wrapper over the adios2 high-level API (v2.9 FileReader
    with a fallback to the stream API)."""

    def __init__(self, filename: str):
        import adios2
        self._mod = adios2
        if hasattr(adios2, "FileReader"):
            self._f = adios2.FileReader(filename)
        else:  # pragma: no cover - legacy adios2
            self._f = adios2.open(filename, "r")

    def close(self):
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()

    def attribute_names(self) -> List[str]:
        av = self._f.available_attributes()
        return list(av.keys())

    def read_attribute_string(self, name: str) -> List[str]:
        v = self._f.read_attribute_string(name)
        return list(v) if isinstance(v, (list, tuple)) else [v]

    def read_attribute(self, name: str) -> np.ndarray:
        return np.asarray(self._f.read_attribute(name))

    def read(self, name: str, start: Optional[Sequence[int]] = None,
             count: Optional[Sequence[int]] = None) -> np.ndarray:
        if start is None:
            return np.asarray(self._f.read(name))
        return np.asarray(self._f.read(name, start=list(start),
                                       count=list(count)))


_TORCH_LONG_KEYS = {"edge_index", "z", "atomic_numbers", "dataset_name"}


def read_adios_samples(f, label: str,
                       keys: Optional[Sequence[str]] = None
                       ) -> List[Data]:
    """Read every sample of ``label`` from an open ``_Adios2File``
    (or API-compatible mock)."""
    attr_names = set(f.attribute_names())
    if f"{label}/keys" not in attr_names:
        raise KeyError(f"label '{label}' not present")
    all_keys = f.read_attribute_string(f"{label}/keys")
    if keys is not None:
        all_keys = [k for k in all_keys if k in set(keys)]
    ndata = int(f.read_attribute(f"{label}/ndata").reshape(-1)[0])

    arrays: Dict[str, np.ndarray] = {}
    counts: Dict[str, np.ndarray] = {}
    offsets: Dict[str, np.ndarray] = {}
    vdims: Dict[str, int] = {}
    for k in all_keys:
        if k == "dataset_name":
            continue
        arrays[k] = f.read(f"{label}/{k}")
        counts[k] = f.read(f"{label}/{k}/variable_count").reshape(-1)
        offsets[k] = f.read(
            f"{label}/{k}/variable_offset").reshape(-1)
        vd = f.read(f"{label}/{k}/variable_dim")
        vdims[k] = int(np.asarray(vd).reshape(-1)[0])

    samples: List[Data] = []
    for i in range(ndata):
        d = Data()
        for k, arr in arrays.items():
            off = int(offsets[k][i])
            cnt = int(counts[k][i])
            vdim = vdims[k]
            sl = [slice(None)] * arr.ndim
            sl[vdim] = slice(off, off + cnt)
            a = np.ascontiguousarray(arr[tuple(sl)])
            t = torch.from_numpy(a.copy())
            if k in _TORCH_LONG_KEYS or np.issubdtype(a.dtype,
                                                      np.integer):
                t = t.long()
            else:
                t = t.float()
            d[k] = t
        samples.append(d)
    return samples


def read_global_attributes(f) -> Dict[str, object]:
    out: Dict[str, object] = {}
    names = set(f.attribute_names())
    for name in ("minmax_graph_feature", "minmax_node_feature",
                 "pna_deg"):
        if name in names:
            out[name] = f.read_attribute(name)
    if "dataset_name" in names:
        v = f.read_attribute_string("dataset_name")
        out["dataset_name"] = v[0] if v else None
    return out


def convert_bp_to_graphstore(bp_path: str, out_dir: str,
                             labels: Sequence[str] = LABELS,
                             file_cls=None) -> Dict[str, int]:
    """Convert a reference ``.bp`` store into the native GraphStore
    layout (one store per label).  Returns {label: num_samples}."""
    from .graphstore import GraphStoreWriter

    file_cls = file_cls or _Adios2File
    os.makedirs(out_dir, exist_ok=True)
    converted: Dict[str, int] = {}
    with file_cls(bp_path) as f:
        attrs = read_global_attributes(f)
        for label in labels:
            try:
                samples = read_adios_samples(f, label)
            except KeyError:
                continue
            w = GraphStoreWriter(label, out_dir)
            w.add(samples)
            for k, v in attrs.items():
                if v is not None:
                    w.add_global(k, np.asarray(v).tolist()
                                 if isinstance(v, np.ndarray) else v)
            w.save()
            converted[label] = len(samples)
    return converted


def main(argv=None):  # pragma: no cover - CLI shell
    ap = argparse.ArgumentParser(
        description="Convert a reference HydraGNN ADIOS2 .bp dataset "
                    "into the native GraphStore format")
    ap.add_argument("--bp", required=True)
    ap.add_argument("--out", required=True)
    ap.add_argument("--labels", nargs="*", default=list(LABELS))
    args = ap.parse_args(argv)
    converted = convert_bp_to_graphstore(args.bp, args.out, args.labels)
    for label, n in converted.items():
        print(f"{label}: {n} samples")
    if not converted:
        print("no labels found")


if __name__ == "__main__":  # pragma: no cover
    main()
