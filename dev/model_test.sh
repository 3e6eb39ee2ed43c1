#!/usr/bin/env bash
# multi-process model topology tests (gloo CPU; reference: dev/model_test.sh
# launched 4-GPU jobs -- here the same oracles run as 2-process gloo suites)
set -e
cd "$(dirname "$0")/.."
python -m pytest tests/parallel tests/test_trainer.py -q "$@"
