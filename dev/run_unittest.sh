#!/usr/bin/env bash
# CPU unit tests (reference: dev/run_unittest.sh)
set -e
cd "$(dirname "$0")/.."
python -m pytest tests -q -m "not gpu" "$@"
