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


class AdiosDataset(torch.utils.data.Dataset):
    """Map-style dataset over one label of a reference ``.bp`` store
    (reference adiosdataset.py:365-790 AdiosDataset, preload mode).
    Samples load eagerly at construction; global attributes
    (``pna_deg``, minmax features, ``dataset_name``) are exposed as
    attributes for update_config."""

    def __init__(self, filename: str, label: str = "trainset",
                 keys: Optional[Sequence[str]] = None, file_cls=None):
        file_cls = file_cls or _Adios2File
        with file_cls(filename) as f:
            self.samples = read_adios_samples(f, label, keys=keys)
            attrs = read_global_attributes(f)
        self.label = label
        self.pna_deg = attrs.get("pna_deg")
        self.minmax_node_feature = attrs.get("minmax_node_feature")
        self.minmax_graph_feature = attrs.get("minmax_graph_feature")
        self.dataset_name = attrs.get("dataset_name")

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        return self.samples[idx]


class AdiosMultiDataset(torch.utils.data.Dataset):
    """Concatenation of one label across several ``.bp`` stores
    (reference adiosdataset.py:1128 AdiosMultiDataset)."""

    def __init__(self, filenames: Sequence[str], label: str = "trainset",
                 keys: Optional[Sequence[str]] = None, file_cls=None):
        self.datasets = [AdiosDataset(fn, label, keys=keys,
                                      file_cls=file_cls)
                         for fn in filenames]
        self._lens = [len(d) for d in self.datasets]
        self.pna_deg = next((d.pna_deg for d in self.datasets
                             if d.pna_deg is not None), None)

    def __len__(self):
        return sum(self._lens)

    def __getitem__(self, idx):
        for d, n in zip(self.datasets, self._lens):
            if idx < n:
                return d[idx]
            idx -= n
        raise IndexError(idx)


class AdiosWriter:
    """Write datasets in the reference ``.bp`` layout (reference
    adiosdataset.py:120-287 AdiosWriter): per label a concatenated
    global array per key plus variable_count/offset/dim metadata, and
    global attributes.  The IO backend is injectable for testing; the
    default uses the adios2 high-level Stream API."""

    def __init__(self, filename: str, backend=None):
        self.filename = filename
        self.backend = backend
        self._labels: Dict[str, List[Data]] = {}
        self.attributes: Dict[str, object] = {}

    def add_global(self, name: str, value) -> None:
        self.attributes[name] = value

    def add(self, label: str, samples: Sequence[Data]) -> None:
        self._labels.setdefault(label, []).extend(samples)

    @staticmethod
    def _pack(samples: Sequence[Data], key: str):
        """Concatenate one key across samples along its variable dim
        (dim 1 for edge_index-style [fixed, n] tensors, else dim 0)
        with count/offset metadata."""
        arrs = [np.asarray(s[key].detach().cpu().numpy())
                for s in samples]
        vdim = 1 if (arrs[0].ndim == 2 and key == "edge_index") else 0
        counts = np.array([a.shape[vdim] for a in arrs], dtype=np.int64)
        offsets = np.concatenate([[0], np.cumsum(counts)[:-1]])
        return np.concatenate(arrs, axis=vdim), counts, offsets, vdim

    def save(self) -> None:
        backend = self.backend
        if backend is None:
            backend = _Adios2StreamBackend(self.filename)
        with backend:
            for name, value in self.attributes.items():
                backend.write_attribute(name, value)
            for label, samples in self._labels.items():
                if not samples:
                    continue
                keys = [k for k in samples[0].keys()
                        if torch.is_tensor(samples[0][k])]
                backend.write_attribute(f"{label}/keys", keys)
                backend.write_attribute(f"{label}/ndata",
                                        np.int64(len(samples)))
                for k in keys:
                    arr, counts, offsets, vdim = self._pack(samples, k)
                    backend.write_array(f"{label}/{k}", arr)
                    backend.write_array(f"{label}/{k}/variable_count",
                                        counts)
                    backend.write_array(f"{label}/{k}/variable_offset",
                                        offsets)
                    backend.write_array(f"{label}/{k}/variable_dim",
                                        np.array([vdim], dtype=np.int64))


class _Adios2StreamBackend:  # pragma: no cover - needs adios2
    """Default AdiosWriter backend over adios2's high-level API."""

    def __init__(self, filename: str):
        import adios2
        self._stream = adios2.Stream(filename, "w")

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self._stream.close()

    def write_attribute(self, name, value):
        if isinstance(value, (list, tuple)) and value and \
                isinstance(value[0], str):
            self._stream.write_attribute(name, list(value))
        else:
            self._stream.write_attribute(name, np.asarray(value))

    def write_array(self, name, arr):
        arr = np.ascontiguousarray(arr)
        self._stream.write(name, arr, shape=list(arr.shape),
                           start=[0] * arr.ndim,
                           count=list(arr.shape))
