#!/bin/bash
# Longformer MLM demo at 1k context (tiny random-init unless --model_path).
set -euo pipefail
cd "$(dirname "$0")"
exec python longformer_mlm.py "$@"
