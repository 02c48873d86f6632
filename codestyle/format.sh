#!/usr/bin/env bash
# format C++/HIP sources in-place
cd "$(dirname "$0")/.."
command -v clang-format >/dev/null && clang-format -i csrc/*.cpp csrc/*.h csrc/*.hip
command -v flake8 >/dev/null && flake8 --config codestyle/flake8.cfg paddlefleetx_amd tools || true
