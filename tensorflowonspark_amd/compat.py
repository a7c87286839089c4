"""Runtime-version shims (parity: reference ``compat.py``, which papered over
TF1/TF2 differences). The torch equivalents are stable; these exist so
reference user code ports line-for-line."""

from . import gpu_info


def export_saved_model(model, export_dir, is_chief=False):
    """Chief-only TorchScript export (reference ``compat.py:10-17`` had
    non-chief ranks write to a dummy path; here they simply skip)."""
    from . import TFNode
    return TFNode.export_saved_model(model, export_dir, is_chief=is_chief)


def disable_auto_shard(options):
    """No-op: torch DataLoader/DataFeed sharding is explicit per-rank."""
    return options


def is_gpu_available():
    return gpu_info.is_gpu_available()
