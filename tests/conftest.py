import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (gfx950) GPU")


def _ensure_built():
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    orcl = os.path.join(root, "oracle", "liborcl.so")
    prod = os.path.join(root, "yugabyte-db_amd", "libybgpu.so")
    if not (os.path.exists(orcl) and os.path.exists(prod)):
        subprocess.run([sys.executable, "-c",
                        "import __graft_entry__; __graft_entry__.build()"],
                       cwd=root, check=True)


_ensure_built()
