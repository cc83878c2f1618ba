#!/bin/bash
# Randeng reasoning generation demo (tiny random-init unless --model_path).
set -e
cd "$(dirname "$0")/../.."
exec python examples/randeng_reasoning/reasoning_generate.py "$@"
