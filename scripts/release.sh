#!/usr/bin/env bash
# Release build (the reference .goreleaser.yaml analog): versioned
# container images for the two targets + an sdist of the package.
# Usage: ./scripts/release.sh v0.2.0 [registry-prefix]
set -euo pipefail
VERSION=${1:?usage: release.sh vX.Y.Z [registry]}
REG=${2:-ghcr.io/example/inference-gateway-amd}

grep -q "${VERSION#v}" inference_gateway_amd/__init__.py || {
  echo "version ${VERSION#v} not found in inference_gateway_amd/__init__.py" >&2
  exit 1
}
python -m pytest tests -q -m "not gpu"
python tools/generate.py
git diff --exit-code || {
  echo "generated artifacts drifted — commit tools/generate.py output first" >&2
  exit 1
}
docker build -t "$REG:$VERSION" -t "$REG:latest" --target gateway .
docker build -t "$REG:$VERSION-engine" --target engine .
python setup.py sdist
echo "built: $REG:$VERSION (+ -engine), dist/$(ls dist | tail -1)"
echo "push with: docker push $REG:$VERSION && docker push $REG:$VERSION-engine"
