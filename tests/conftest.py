import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that require an MI355X GPU (run via gpurun)"
    )
    config.addinivalue_line(
        "markers", "multigpu: tests that require more than one GPU"
    )


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords or "multigpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture()
def tmp_db(tmp_path):
    """A CPU-engine SwarmsDB saving under tmp_path."""
    from swarmdb_amd import QueueConfig, SwarmsDB

    cfg = QueueConfig(use_gpu=False, save_dir=str(tmp_path / "history"),
                      max_agents=512)
    db = SwarmsDB(config=cfg)
    yield db
    db.config.auto_save = False
    db.close()
