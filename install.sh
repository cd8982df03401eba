#!/bin/sh
# Install the MI355X-native inference gateway from a source checkout
# (the analog of the reference's install.sh binary installer).
set -e
PREFIX="${PREFIX:-/usr/local}"
PYTHON="${PYTHON:-python3}"

echo "==> checking python"
$PYTHON -c 'import sys; assert sys.version_info >= (3, 10)'

echo "==> installing python dependencies"
$PYTHON -m pip install --quiet httpx uvicorn pyyaml

if $PYTHON -c 'import torch' 2>/dev/null && \
   $PYTHON -c 'import torch; assert torch.version.hip' 2>/dev/null; then
  echo "==> ROCm PyTorch detected: building the gfx950 HIP extension"
  PYTORCH_ROCM_ARCH=gfx950 $PYTHON setup.py build_ext --inplace
else
  echo "==> no ROCm PyTorch: gateway-only install (local engine disabled)"
fi

BIN="$PREFIX/bin/inference-gateway-amd"
echo "==> writing $BIN"
cat > "$BIN" <<WRAP
#!/bin/sh
exec $PYTHON -m inference_gateway_amd "\$@"
WRAP
chmod +x "$BIN"
echo "==> done; run: inference-gateway-amd"
