#!/bin/sh
# Enforce generated-artifact cleanliness + quick tests before committing
# (analog of the reference's scripts/pre-commit-check.sh).
set -e
python tools/generate.py
if ! git diff --quiet -- Configurations.md .env.example \
    inference_gateway_amd/providers/constants_gen.py; then
  echo "generated artifacts drifted: commit the regenerated files" >&2
  exit 1
fi
python -m pytest tests -m "not gpu" -q
