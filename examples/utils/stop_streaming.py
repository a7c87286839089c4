#!/usr/bin/env python3
"""Signal a running streaming cluster to stop feeding
(parity: reference ``examples/utils/stop_streaming.py:12-18`` — connect to the
reservation server and send STOP)."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

from tensorflowonspark_amd import reservation  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("host")
    p.add_argument("port", type=int)
    args = p.parse_args()
    client = reservation.Client((args.host, args.port))
    client.request_stop()
    client.close()
    print("stop requested at {}:{}".format(args.host, args.port))


if __name__ == "__main__":
    main()
