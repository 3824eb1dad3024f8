#!/usr/bin/env python3
"""Reference-parity launcher: `torchrun --nproc_per_node=N
main.py --preset ...` or single-process `python main.py ...`."""

import sys

from sodkit.cli import main

if __name__ == "__main__":
    sys.exit(main())
