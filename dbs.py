#!/usr/bin/env python3
"""dbs.py — CLI entry point (frozen API surface of the reference's dbs.py).

Same 13 flags, defaults, artifact layout (./logs, ./statis) and
skip-if-already-done guard (reference dbs.py:527-544); execution is the
MI355X-native framework in dynamic_load_balance_distributeddnn_amd/.
"""

import os
import sys

from dynamic_load_balance_distributeddnn_amd.cli import base_filename, get_parser
from dynamic_load_balance_distributeddnn_amd.launcher import launch


def main(argv=None) -> int:
    args = get_parser().parse_args(argv)
    name = base_filename(args)
    # Idempotent experiment harness: skip a finished run (dbs.py:528-534).
    if os.path.isfile(os.path.join("./logs", name.format("0") + ".log")):
        print("\n===========================")
        print("Experiment already finished, skipping...")
        print("===========================\n")
        return 0
    launch(args)
    return 0


if __name__ == "__main__":
    sys.exit(main())
