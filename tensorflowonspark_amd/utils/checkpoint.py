"""Checkpoint layout helpers (conventions parity: survey §5 checkpoint/resume).

``model_dir`` holds per-epoch/step checkpoints named ``weights-NNNN.pt``
(reference Keras convention ``model_dir/weights-{epoch:04d}``,
``mnist_spark.py:51-53``); ``export_dir`` holds the serving export written by
``TFNode.export_saved_model`` (chief-only). Resume = point the same
``model_dir`` at a rerun and call ``load_latest``.
"""

import glob
import logging
import os
import re

logger = logging.getLogger(__name__)

_PAT = re.compile(r"weights-(\d+)\.pt$")


def checkpoint_path(model_dir, step):
    return os.path.join(model_dir, "weights-{:04d}.pt".format(step))


def save_checkpoint(model_dir, step, model, optimizer_state=None, extra=None,
                    keep_last=5):
    """Save model (+ optional optimizer state) as ``model_dir/weights-NNNN.pt``."""
    import torch
    os.makedirs(model_dir, exist_ok=True)
    payload = {"step": step, "model": model.state_dict()}
    if optimizer_state is not None:
        payload["optimizer"] = optimizer_state
    if extra:
        payload["extra"] = extra
    path = checkpoint_path(model_dir, step)
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)  # atomic: a crash never leaves a torn checkpoint
    if keep_last:
        # sort by numeric step (lexicographic breaks past the 4-digit padding:
        # 'weights-10000.pt' < 'weights-9999.pt')
        ckpts = [p for p in glob.glob(os.path.join(model_dir, "weights-*.pt"))
                 if _PAT.search(p)]
        ckpts.sort(key=lambda p: int(_PAT.search(p).group(1)))
        for old in ckpts[:-keep_last]:
            try:
                os.remove(old)
            except OSError:
                pass
    logger.info("saved checkpoint %s", path)
    return path


def latest_checkpoint(model_dir):
    """Path of the newest checkpoint in model_dir, or None
    (parity: ``tf.train.latest_checkpoint``, reference ``pipeline.py:549-555``)."""
    best, best_step = None, -1
    for path in glob.glob(os.path.join(model_dir, "weights-*.pt")):
        m = _PAT.search(path)
        if m and int(m.group(1)) > best_step:
            best, best_step = path, int(m.group(1))
    return best


def load_checkpoint(path, model, map_location="cpu"):
    """Load a checkpoint; returns (step, optimizer_state_or_None)."""
    import torch
    payload = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(payload["model"])
    logger.info("restored %s (step %s)", path, payload.get("step"))
    return payload.get("step", 0), payload.get("optimizer")


def load_latest(model_dir, model, map_location="cpu"):
    """Resume from the newest checkpoint if one exists; returns step (0 if none)."""
    path = latest_checkpoint(model_dir)
    if path is None:
        return 0, None
    return load_checkpoint(path, model, map_location)
