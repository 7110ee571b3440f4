import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")


@pytest.fixture()
def store(tmp_path):
    from agentainer_amd.store import Store

    s = Store(str(tmp_path / "state"), sync="interval")
    yield s
    s.close()


@pytest.fixture()
def runtime(tmp_path):
    """Runtime over the echo engine with fast worker intervals, no threads."""
    from agentainer_amd.config import load_config
    from agentainer_amd.engine.echo import EchoEngine
    from agentainer_amd.service import Runtime
    from agentainer_amd.store import Store

    cfg = load_config(path="/nonexistent.yaml", env={})
    cfg.data["store"]["path"] = str(tmp_path / "root")
    s = Store(str(tmp_path / "root" / "state"), sync="interval")
    rt = Runtime(cfg, engine=EchoEngine(s), store=s, state_root=str(tmp_path / "root"))
    yield rt
    rt.shutdown()
