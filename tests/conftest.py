import sys
from pathlib import Path

import pytest

# repo root on sys.path so `import pertgnn` works without installation
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)"
    )


@pytest.fixture(scope="session")
def synthetic_workspace(tmp_path_factory):
    """Small synthetic dataset written to disk + ingested artifacts."""
    from pertgnn.data.ingest import run_ingest
    from pertgnn.data.synthetic import SyntheticConfig, write_dataset

    root = tmp_path_factory.mktemp("synth")
    cfg = SyntheticConfig(n_entries=3, patterns_per_entry=2, traces_per_entry=30,
                          min_calls=3, max_calls=8, n_microservices=24, seed=7)
    write_dataset(str(root), cfg)
    out = run_ingest(
        data_root=str(root / "data"),
        processed_dir=str(root / "processed"),
        min_occurence=10,
        verbose=False,
    )
    return root, out
