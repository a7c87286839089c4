"""Small host utilities (parity: reference ``tensorflowonspark/util.py``)."""

import errno
import logging
import os
import socket

from . import gpu_info

logger = logging.getLogger(__name__)


def single_node_env(num_gpus=1, worker_index=-1, nodes=None):
    """Set up environment for a standalone (non-cluster) instance.

    MI355X note: GPU visibility is exported as both ``HIP_VISIBLE_DEVICES`` and
    ``CUDA_VISIBLE_DEVICES`` (PyTorch-ROCm honors the latter); reference
    equivalent at ``util.py:21-49`` used CUDA only.
    """
    if num_gpus and gpu_info.is_gpu_available():
        gpus = gpu_info.get_gpus(num_gpus, worker_index)
        gpu_str = gpus if isinstance(gpus, str) else ",".join(gpus)
        os.environ["HIP_VISIBLE_DEVICES"] = gpu_str
        os.environ["CUDA_VISIBLE_DEVICES"] = gpu_str
    else:
        os.environ.setdefault("HIP_VISIBLE_DEVICES", "")
        os.environ.setdefault("CUDA_VISIBLE_DEVICES", "")


def get_ip_address():
    """Externally-facing IP of this host (UDP connect trick, no packet sent)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("8.8.8.8", 53))
        return s.getsockname()[0]
    except OSError:
        return "127.0.0.1"
    finally:
        s.close()


def find_in_path(path, file_name):
    """Find a file in a colon-separated path string."""
    for p in path.split(os.pathsep):
        candidate = os.path.join(p, file_name)
        if os.path.exists(candidate) and os.path.isfile(candidate):
            return candidate
    return False


EXECUTOR_ID_FILE = "executor_id"


def write_executor_id(num, cwd=None):
    """Persist this executor's id in its working dir.

    The file is the identity link between the bootstrap task and later
    feeder/shutdown tasks landing on the same executor (reference
    ``util.py:77-94`` semantics).
    """
    path = os.path.join(cwd or os.getcwd(), EXECUTOR_ID_FILE)
    with open(path, "w") as f:
        f.write(str(num))


def read_executor_id(cwd=None):
    path = os.path.join(cwd or os.getcwd(), EXECUTOR_ID_FILE)
    try:
        with open(path) as f:
            return int(f.read())
    except OSError as e:
        if e.errno == errno.ENOENT:
            raise RuntimeError(
                "No executor_id file found in {} — was the cluster started here?".format(
                    os.path.dirname(path)))
        raise
