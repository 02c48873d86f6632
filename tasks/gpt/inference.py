#!/usr/bin/env python3
"""Task entry (reference tasks/gpt/inference.py:35-60) — thin wrapper
over tools/inference.py for surface parity."""
import os
import runpy
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
sys.argv[0] = os.path.join(os.path.dirname(__file__), "..", "..",
                           "tools", "inference.py")
runpy.run_path(sys.argv[0], run_name="__main__")
