"""Example scripts run as subprocesses (pattern: reference
tests/test_examples.py:18-90)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, *args, timeout=420):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    return subprocess.run(
        [sys.executable, os.path.join(REPO, script), *args],
        capture_output=True, text=True, timeout=timeout, cwd=REPO,
        env=env)


@pytest.mark.parametrize("mpnn_type", ["PNA", "SchNet"])
def test_qm9_example(mpnn_type, tmp_path):
    r = _run("examples/qm9/qm9.py", "--mpnn_type", mpnn_type,
             "--num_epoch", "2", "--num_samples", "48")
    assert r.returncode == 0, r.stderr[-2000:]


@pytest.mark.parametrize("mpnn_type", ["MACE", "EGNN"])
def test_md17_mlip_example(mpnn_type):
    r = _run("examples/md17/md17_mlip.py", "--mpnn_type", mpnn_type,
             "--num_epoch", "2", "--num_samples", "24")
    assert r.returncode == 0, r.stderr[-2000:]


def test_lennardjones_example():
    r = _run("examples/LennardJones/LennardJones.py",
             "--num_epoch", "2", "--num_samples", "16")
    assert r.returncode == 0, r.stderr[-2000:]


def test_multibranch_example():
    r = _run("examples/multibranch/train.py", "--num_epoch", "2")
    assert r.returncode == 0, r.stderr[-2000:]


def test_multidataset_example():
    r = _run("examples/multidataset/train.py", "--num_epoch", "2")
    assert r.returncode == 0, r.stderr[-2000:]


def test_ising_example():
    r = _run("examples/ising/ising.py", "--num_epoch", "2")
    assert r.returncode == 0, r.stderr[-2000:]


def test_eam_example():
    r = _run("examples/eam/eam.py", "--num_epoch", "2")
    assert r.returncode == 0, r.stderr[-2000:]


def test_lsms_example():
    r = _run("examples/lsms/lsms.py", "--num_epoch", "2")
    assert r.returncode == 0, r.stderr[-2000:]


def test_open_catalyst_example():
    r = _run("examples/open_catalyst/open_catalyst.py",
             "--num_epoch", "2", "--num_samples", "12")
    assert r.returncode == 0, r.stderr[-2000:]


def test_qm9_hpo_example():
    r = _run("examples/qm9_hpo/qm9_hpo.py", "--trials", "2",
             "--epochs", "1", "--samples", "48")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "best params" in r.stdout


NEW_EXAMPLES = [
    "ani1_x/ani1_x.py",
    "transition1x/transition1x.py",
    "alexandria/alexandria.py",
    "mptrj/mptrj.py",
    "open_catalyst_2020/open_catalyst_2020.py",
    "open_catalyst_2022/open_catalyst_2022.py",
    "open_catalyst_2025/open_catalyst_2025.py",
    "open_direct_air_capture_2023/odac2023.py",
    "open_materials_2024/open_materials_2024.py",
    "open_molecules_2025/open_molecules_2025.py",
    "open_polymers_2026/open_polymers_2026.py",
    "qm7x/qm7x.py",
    "nabla2_dft/nabla2_dft.py",
    "qcml/qcml.py",
    "csce/csce.py",
    "zinc/zinc.py",
    "ogb/ogb.py",
]


@pytest.mark.parametrize("script", NEW_EXAMPLES,
                         ids=[s.split("/")[0] for s in NEW_EXAMPLES])
def test_reference_parity_examples(script):
    """Every reference example dir has a runnable counterpart
    (reference examples/* — 31 dirs); synthetic data of the same
    shape, same config-driven flow."""
    r = _run(f"examples/{script}", "--num_epoch", "2",
             "--num_samples", "10")
    assert r.returncode == 0, r.stderr[-2000:]


def test_dftb_uv_spectrum_discrete():
    r = _run("examples/dftb_uv_spectrum/dftb_uv_spectrum.py",
             "--num_epoch", "2", "--num_samples", "10",
             "--mode", "discrete")
    assert r.returncode == 0, r.stderr[-2000:]


@pytest.mark.parametrize("script", [
    "multibranch_hpo/train.py", "multidataset_hpo/train.py",
    "multidataset_hpo_sc26/train.py",
], ids=["multibranch_hpo", "multidataset_hpo", "sc26"])
def test_hpo_flow_examples(script):
    r = _run(f"examples/{script}", "--trials", "1", "--num_epoch", "1",
             timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "best params" in r.stdout


def test_multidataset_deepspeed_example():
    r = _run("examples/multidataset_deepspeed/train.py",
             "--num_epoch", "1")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "DDP fallback" in r.stdout or "deepspeed engine" in r.stdout
