#!/bin/sh
# Hardware numerics suite — run on an MI355X node.
set -e
python -m pytest tests -q -m gpu "$@"
