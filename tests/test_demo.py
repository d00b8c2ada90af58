"""End-to-end demo test (reference test_demo_node.py pattern): worker pool
+ client MAP optimization recover the true parameters."""
import multiprocessing
import socket
import time

import numpy as np
import pytest

DEMO_PORTS = (9541, 9542)


def _wait_tcp(port, timeout=30.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return
        except OSError:
            time.sleep(0.1)
    raise TimeoutError(f"port {port} never opened")


def _node(port, seed):
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
    from demo_node import run_node

    run_node("127.0.0.1", port, 0.0, "cpu", 60, seed)


@pytest.mark.timeout(300)
def test_demo_map_recovers_truth():
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
    from demo_model import run_model

    ctx = multiprocessing.get_context("spawn")
    procs = [ctx.Process(target=_node, args=(p, i), daemon=True) for i, p in enumerate(DEMO_PORTS)]
    for p in procs:
        p.start()
    try:
        for port in DEMO_PORTS:
            _wait_tcp(port)
        chain = run_model("127.0.0.1", list(DEMO_PORTS), parallel=True, map_steps=120, draws=60)
        # ground truth of generate_linear_dataset: intercept 1.5, slope 0.5
        assert abs(chain[:, 0].mean() - 1.5) < 0.3
        assert abs(chain[:, 1].mean() - 0.5) < 0.1
        # NUTS over the same live workers (each leapfrog = one federated call)
        chain = run_model("127.0.0.1", list(DEMO_PORTS), parallel=True,
                          map_steps=0, draws=40, sampler="nuts")
        assert abs(chain[:, 0].mean() - 1.5) < 0.3
        assert abs(chain[:, 1].mean() - 0.5) < 0.1
    finally:
        for p in procs:
            if p.is_alive():
                p.terminate()
        for p in procs:
            p.join(timeout=10)
