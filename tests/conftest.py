import sys
from pathlib import Path

# allow `import wva_amd` (repo root) and `import vllm_emulator` (tools/)
# from a source checkout without installation
_ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(_ROOT))
sys.path.insert(0, str(_ROOT / "tools"))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD MI355X GPU (run via gpurun)")
