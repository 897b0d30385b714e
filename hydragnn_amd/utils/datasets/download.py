"""Resumable, checksum-verified dataset downloads + traversal-safe tar
extraction (reference: hydragnn/utils/datasets/download.py:23-113;
CLI: python -m hydragnn_amd.utils.datasets.download URL DEST)."""

from __future__ import annotations

import hashlib
import os
import sys
import tarfile
import urllib.request
from typing import Optional


def sha256_of(path: str) -> str:
    h = hashlib.sha256()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()


def download(url: str, dest: str, sha256: Optional[str] = None,
             chunk_size: int = 1 << 20) -> str:
    """Resumable download via HTTP Range into dest + '.part', renamed
    on success; verifies sha256 when given."""
    os.makedirs(os.path.dirname(os.path.abspath(dest)), exist_ok=True)
    if os.path.exists(dest):
        if sha256 is None or sha256_of(dest) == sha256:
            return dest
        os.remove(dest)
    part = dest + ".part"
    start = os.path.getsize(part) if os.path.exists(part) else 0
    req = urllib.request.Request(url)
    if start:
        req.add_header("Range", f"bytes={start}-")
    mode = "ab" if start else "wb"
    with urllib.request.urlopen(req, timeout=60) as resp:
        if start and resp.status == 200:
            # server ignored Range: restart
            start = 0
            mode = "wb"
        with open(part, mode) as f:
            while True:
                chunk = resp.read(chunk_size)
                if not chunk:
                    break
                f.write(chunk)
    if sha256 is not None and sha256_of(part) != sha256:
        raise ValueError(f"sha256 mismatch for {url}")
    os.replace(part, dest)
    return dest


def safe_extract_tar(tar_path: str, dest_dir: str) -> None:
    """Extract refusing path traversal (../ or absolute members)."""
    dest_dir = os.path.abspath(dest_dir)
    with tarfile.open(tar_path) as tar:
        for member in tar.getmembers():
            target = os.path.abspath(os.path.join(dest_dir, member.name))
            if not target.startswith(dest_dir + os.sep) and \
                    target != dest_dir:
                raise ValueError(
                    f"unsafe tar member path: {member.name}")
            if member.issym() or member.islnk():
                link_target = os.path.abspath(os.path.join(
                    os.path.dirname(target), member.linkname))
                if not link_target.startswith(dest_dir):
                    raise ValueError(
                        f"unsafe tar link: {member.name}")
        tar.extractall(dest_dir)


def main(argv=None):
    argv = argv if argv is not None else sys.argv[1:]
    if len(argv) < 2:
        print("usage: python -m hydragnn_amd.utils.datasets.download "
              "URL DEST [SHA256]")
        return 1
    url, dest = argv[0], argv[1]
    sha = argv[2] if len(argv) > 2 else None
    download(url, dest, sha)
    if dest.endswith((".tar", ".tar.gz", ".tgz")):
        safe_extract_tar(dest, os.path.dirname(dest) or ".")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())


# reference-named alias
download_file = download
