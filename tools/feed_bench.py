#!/usr/bin/env python3
"""Standalone InputMode.SPARK ingest-path throughput: feeder -> shm ring ->
pinned staging -> (async H2D when a GPU is present)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from bench import SparkFeed  # noqa: E402


def main():
    batch = int(sys.argv[1]) if len(sys.argv) > 1 else 1024
    shape = (224, 224, 3)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    import numpy as np
    slot_bytes = int(np.prod((batch,) + shape)) + batch * 8 + (1 << 16)
    feed = SparkFeed(shape, batch, 1000, device, slot_bytes)
    for _ in range(5):
        feed.next()
    if device.type == "cuda":
        torch.cuda.synchronize()
    n = 40
    t0 = time.time()
    for _ in range(n):
        x, y = feed.next()
    if device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.time() - t0
    gb = n * slot_bytes / 1e9
    print("feed path: {:.2f} GB/s, {:.0f} images/s (batch {}, {} blocks)".format(
        gb / dt, n * batch / dt, batch, n))
    feed.close()


if __name__ == "__main__":
    main()
