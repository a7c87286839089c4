#!/usr/bin/env python3
"""CSV row -> JSON tensor reshaper (parity: reference
``examples/utils/mnist_reshape.py`` — turns one 'label,784 pixels' CSV line
into the JSON input the serving endpoint / tfosr_infer CLI expects).

    head -1 mnist.csv | python examples/utils/mnist_reshape.py
"""

import json
import sys


def reshape_line(line, side=28):
    vals = [float(v) for v in line.strip().split(",")]
    label, pixels = int(vals[0]), vals[1:]
    assert len(pixels) == side * side, len(pixels)
    image = [[pixels[r * side + c] / 255.0 for c in range(side)]
             for r in range(side)]
    return {"label": label, "image": [image]}  # [1, 28, 28]


if __name__ == "__main__":
    for ln in sys.stdin:
        if ln.strip():
            print(json.dumps(reshape_line(ln)))
