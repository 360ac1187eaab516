import os
import socket
import sys
from pathlib import Path

import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))

# torch.distributed rendezvous must use loopback (container hostname may not
# resolve)
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")

# Spark layer tests run on the real pyspark when installed, else on the
# bundled JVM-free localspark runtime — either way they RUN, not skip.
from sparkflow_amd import localspark as _localspark  # noqa: E402

_localspark.install(quiet=True)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)"
    )


def free_port() -> int:
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def tcp_port():
    return free_port()
