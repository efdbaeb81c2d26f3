#!/bin/sh
# Host-side suite (no GPU needed) — reference test_local.sh analog.
set -e
python -m pytest tests -q -m "not gpu" "$@"
