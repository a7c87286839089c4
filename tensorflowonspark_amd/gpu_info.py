"""MI355X GPU discovery and allocation via ``amd-smi``/``rocm-smi``.

Re-targets the capability of reference ``tensorflowonspark/gpu_info.py`` (which
parsed ``nvidia-smi``): probe the node's GPUs, find free ones (no compute process
attached), and hand each worker a deterministic slice.

Placement math parity (reference ``gpu_info.py:80-91``): with ``worker_index==-1``
the free list is randomly shuffled; otherwise worker *i* takes slice
``[i*num_gpu : (i+1)*num_gpu]`` of the free list with modulo wraparound, so
co-located workers get disjoint GPUs.
"""

import json
import logging
import random
import subprocess
import time

logger = logging.getLogger(__name__)

MAX_RETRIES = 3


def _run(cmd):
    """Run a command, return stdout str or None. Injectable for tests."""
    try:
        out = subprocess.run(cmd, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                             timeout=60, check=False)
        if out.returncode != 0:
            return None
        return out.stdout.decode("utf-8", errors="replace")
    except (OSError, subprocess.TimeoutExpired):
        return None


def _list_gpu_ids():
    """Return the list of GPU indices on this node (ints), [] if none/unknown."""
    # amd-smi (ROCm >= 6) JSON listing
    out = _run(["amd-smi", "list", "--json"])
    if out:
        try:
            data = json.loads(out)
            # amd-smi emits either a list of {"gpu": N, ...} or {"gpu_N": {...}}
            if isinstance(data, list):
                ids = [int(entry["gpu"]) for entry in data if "gpu" in entry]
                if ids:
                    return sorted(ids)
            elif isinstance(data, dict):
                ids = []
                for k in data:
                    if k.startswith("gpu"):
                        try:
                            ids.append(int(k.split("_")[-1].replace("gpu", "") or 0))
                        except ValueError:
                            pass
                if ids:
                    return sorted(ids)
        except (ValueError, KeyError, TypeError):
            pass
    # rocm-smi fallback
    out = _run(["rocm-smi", "--showid", "--json"])
    if out:
        try:
            data = json.loads(out)
            ids = []
            for k in data:  # keys like "card0"
                if k.startswith("card"):
                    try:
                        ids.append(int(k[4:]))
                    except ValueError:
                        pass
            return sorted(ids)
        except ValueError:
            pass
    return []


def _busy_gpu_ids():
    """GPU indices that have a compute process attached."""
    out = _run(["amd-smi", "process", "--json"])
    busy = set()
    if out:
        try:
            data = json.loads(out)
            entries = data if isinstance(data, list) else [data]
            for entry in entries:
                if not isinstance(entry, dict):
                    continue
                procs = entry.get("process_list") or entry.get("processes") or []
                gpu_id = entry.get("gpu")
                if gpu_id is None:
                    continue
                real = []
                for p in procs:
                    if not isinstance(p, dict):
                        continue
                    info = p.get("process_info")
                    if info == "N/A" or info is None:
                        continue
                    # amd-smi emits a placeholder entry with name "N/A" when
                    # no process is attached (observed on MI355X, ROCm 7.2)
                    if isinstance(info, dict) and info.get("name") in ("N/A", None):
                        continue
                    real.append(p)
                if real:
                    busy.add(int(gpu_id))
        except (ValueError, KeyError, TypeError):
            pass
    return busy


def is_gpu_available():
    """True if this node exposes at least one AMD GPU."""
    return len(_list_gpu_ids()) > 0


def get_gpus(num_gpu=1, worker_index=-1, format=list):
    """Allocate ``num_gpu`` free GPUs for this worker.

    Retries up to MAX_RETRIES with 30 s * retry backoff when not enough GPUs are
    free (parity: reference ``gpu_info.py:57-70``). Returns a list of index
    strings, or a comma-joined string when ``format=str``.
    """
    retries = 0
    while True:
        all_ids = _list_gpu_ids()
        busy = _busy_gpu_ids()
        free = [str(i) for i in all_ids if i not in busy]
        logger.info("GPUs: all=%s busy=%s free=%s", all_ids, sorted(busy), free)
        if len(free) >= num_gpu:
            if worker_index == -1:
                random.shuffle(free)
                proposed = free[:num_gpu]
            else:
                # deterministic slice with modulo wraparound for co-located workers
                start = (worker_index * num_gpu) % len(free)
                proposed = [free[(start + i) % len(free)] for i in range(num_gpu)]
            logger.info("worker %d assigned GPUs %s", worker_index, proposed)
            if format == str:
                return ",".join(proposed)
            return proposed
        retries += 1
        if retries > MAX_RETRIES:
            raise RuntimeError(
                "Unable to allocate {} GPUs (free: {})".format(num_gpu, free))
        wait = 30 * retries
        logger.warning("only %d free GPUs, need %d; retry %d in %ds",
                       len(free), num_gpu, retries, wait)
        time.sleep(wait)
