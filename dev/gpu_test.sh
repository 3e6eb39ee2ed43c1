#!/usr/bin/env bash
# MI355X kernel + model tests (run on a GPU box)
set -e
cd "$(dirname "$0")/.."
python -m pytest tests -q -m gpu "$@"
