#!/usr/bin/env bash
# Reference scripts/pre-commit-check.sh analog: generated artifacts must
# be clean and the CPU suite green before a commit.
set -euo pipefail
cd "$(dirname "$0")/.."
python tools/generate.py >/dev/null
if ! git diff --exit-code -- Configurations.md .env.example openapi.yaml \
    inference_gateway_amd/providers/constants_gen.py >/dev/null; then
  echo "generated artifacts drifted — commit tools/generate.py output" >&2
  git diff --stat -- Configurations.md .env.example openapi.yaml \
      inference_gateway_amd/providers/constants_gen.py >&2
  exit 1
fi
python -m pytest tests -q -m "not gpu" -x
