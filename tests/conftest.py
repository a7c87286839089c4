import multiprocessing
import sys
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


# Driver-side torch compute must not spin up OpenMP thread pools: later tests
# fork executor processes from this interpreter, and fork-after-OpenMP
# deadlocks in libgomp. One thread in the test driver keeps forks safe.
try:
    import torch
    torch.set_num_threads(1)
except Exception:
    pass


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on an MI355X box)")


# fork is required: tests exercise multi-process managers/rings whose state is
# inherited at fork time.
try:
    multiprocessing.set_start_method("fork", force=False)
except RuntimeError:
    pass
