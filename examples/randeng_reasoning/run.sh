#!/bin/bash
# Randeng reasoning generation demo (tiny random-init unless --model_path).
set -euo pipefail
cd "$(dirname "$0")"
exec python reasoning_generate.py "$@"
