import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


def free_port() -> str:
    """Fresh ephemeral port for torchrun rendezvous — fixed ports reused
    across runs can race (TIME_WAIT / parallel soaks)."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return str(s.getsockname()[1])


def retry_run(call, attempts=2):
    """Re-invoke a subprocess launcher on transient nonzero exits —
    torchrun agent/rendezvous spawns can race under box load; the caller
    passes a lambda so each attempt draws a fresh free_port()."""
    for _ in range(attempts):
        r = call()
        if r.returncode == 0:
            break
    return r
