#!/usr/bin/env python3
"""ai/train.py — the training entrypoint in the layout the reference README
promises (reference README.md:72-76, ROADMAP.md:126-129: ai/ with models/,
planner/, train.py).  Thin shim over nerrf_amd.train."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from nerrf_amd.train import main  # noqa: E402

if __name__ == "__main__":
    main()
