#!/usr/bin/env python3
"""Generate MNIST-shaped data as CSV and TFRecords.

Parity with reference ``examples/mnist/mnist_data_setup.py`` (which downloaded
MNIST via tfds and wrote ``label,784 ints`` CSV plus TFRecords); this
environment has no network, so the images are synthetic — class-conditional
blobs that a small CNN can actually learn to separate, which keeps the example
end-to-end meaningful.

Usage:
  python examples/mnist/mnist_data_setup.py --output data/mnist --num 6000
"""

import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))


def synthetic_mnist(num, seed=0):
    """Images [num,28,28] uint8, labels [num] — one blob position per class."""
    rng = np.random.default_rng(seed)
    labels = rng.integers(0, 10, size=num)
    images = np.zeros((num, 28, 28), dtype=np.uint8)
    centers = [(7 + 5 * (k % 4), 7 + 5 * (k // 4)) for k in range(10)]
    yy, xx = np.mgrid[0:28, 0:28]
    for i, k in enumerate(labels):
        cy, cx = centers[k]
        blob = np.exp(-(((yy - cy) ** 2 + (xx - cx) ** 2) / 12.0))
        noise = rng.normal(0, 0.08, size=(28, 28))
        images[i] = np.clip((blob + noise) * 255, 0, 255).astype(np.uint8)
    return images, labels


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--output", default="data/mnist")
    p.add_argument("--num", type=int, default=6000)
    p.add_argument("--format", choices=["csv", "tfr", "both"], default="both")
    args = p.parse_args()

    images, labels = synthetic_mnist(args.num)
    os.makedirs(args.output, exist_ok=True)

    if args.format in ("csv", "both"):
        path = os.path.join(args.output, "mnist.csv")
        with open(path, "w") as f:
            for img, lab in zip(images, labels):
                f.write(str(lab) + "," + ",".join(map(str, img.reshape(-1))) + "\n")
        print("wrote", path)

    if args.format in ("tfr", "both"):
        from tensorflowonspark_amd import tfrecord
        tfr_dir = os.path.join(args.output, "tfr")
        os.makedirs(tfr_dir, exist_ok=True)
        per_file = max(1, args.num // 4)
        i = 0
        for part in range(4):
            path = os.path.join(tfr_dir, "part-r-{:05d}".format(part))
            with tfrecord.TFRecordWriter(path) as w:
                for _ in range(per_file):
                    if i >= args.num:
                        break
                    w.write(tfrecord.encode_example({
                        "image": images[i].reshape(-1).astype(np.int64).tolist(),
                        "label": int(labels[i])}))
                    i += 1
        print("wrote", tfr_dir)


if __name__ == "__main__":
    main()
