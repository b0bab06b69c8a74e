import sys
from pathlib import Path

# allow `import wva_amd` from a source checkout without installation
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD MI355X GPU (run via gpurun)")
