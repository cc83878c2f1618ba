#!/bin/bash
# Longformer MLM demo at 1k context (tiny random-init unless --model_path).
set -e
cd "$(dirname "$0")/../.."
exec python examples/longformer/longformer_mlm.py "$@"
