#!/usr/bin/env python3
"""Task entry (reference tasks/gpt/generation.py:35-63) — thin wrapper
over tools/generation.py for surface parity."""
import os
import runpy
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
sys.argv[0] = os.path.join(os.path.dirname(__file__), "..", "..",
                           "tools", "generation.py")
runpy.run_path(sys.argv[0], run_name="__main__")
