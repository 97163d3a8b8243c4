import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
BUILD = REPO / "build"

sys.path.insert(0, str(REPO))
sys.path.insert(0, str(BUILD))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def racon():
    """The _racon native module (builds it if missing)."""
    try:
        import _racon
        return _racon
    except ImportError:
        subprocess.run(
            [sys.executable, str(REPO / "__graft_entry__.py"), "build"], check=True, cwd=REPO
        )
        import _racon
        return _racon


@pytest.fixture(scope="session")
def racon_cli(racon):
    """Path to the racon CLI binary."""
    path = BUILD / "racon"
    assert path.exists()
    return str(path)


@pytest.fixture(scope="session")
def ref_data():
    """Reference sample data dir (only exists in the CPU container)."""
    d = Path("/root/reference/test/data")
    if not d.is_dir():
        pytest.skip("reference sample data not available on this host")
    return d


@pytest.fixture(scope="session")
def sample(tmp_path_factory, racon):
    """Small in-repo synthetic sample (reads, overlaps, layout, reference)."""
    from racon_amd import synth

    d = tmp_path_factory.mktemp("sample")
    return synth.make_sample(d, genome_bp=20000, coverage=20, seed=7)


def read_fasta(path):
    import gzip

    op = gzip.open if str(path).endswith(".gz") else open
    seqs = {}
    name, chunks = None, []
    with op(str(path), "rt") as f:
        for line in f:
            line = line.rstrip()
            if line.startswith(">"):
                if name:
                    seqs[name] = "".join(chunks)
                name, chunks = line[1:].split()[0], []
            else:
                chunks.append(line)
    if name:
        seqs[name] = "".join(chunks)
    return seqs


@pytest.fixture(scope="session")
def fasta_reader():
    return read_fasta
