import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (or any ROCm GPU)")


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
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def tmp_engine(tmp_path):
    from greptimedb_amd.engine.engine import MitoEngine, EngineConfig
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "data"), device="cpu",
                                  background_flush=False))
    yield eng
    eng.close()


@pytest.fixture
def gpu_engine(tmp_path):
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from greptimedb_amd.engine.engine import MitoEngine, EngineConfig
    eng = MitoEngine(EngineConfig(data_dir=str(tmp_path / "data"), device="cuda:0",
                                  background_flush=False))
    yield eng
    eng.close()
