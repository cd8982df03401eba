# MI355X-native inference gateway.
#
# Two build targets (the reference ships a 2-stage static-binary image;
# the Python analog separates the CPU-only gateway from the ROCm engine):
#   gateway  — CPU-only proxy/gateway (remote providers, MCP, guardrails)
#   engine   — ROCm 7.x image with the gfx950 HIP extension built in-tree
#              (requires a ROCm base image + an MI355X at runtime)

FROM python:3.10-slim AS gateway
WORKDIR /app
COPY inference_gateway_amd/ inference_gateway_amd/
COPY spec/ spec/
COPY examples/mcp-servers/ examples/mcp-servers/
RUN pip install --no-cache-dir httpx uvicorn pyyaml
ENV SERVER_HOST=0.0.0.0 SERVER_PORT=8080
EXPOSE 8080 9464
ENTRYPOINT ["python", "-m", "inference_gateway_amd"]

FROM rocm/pytorch:latest AS engine
WORKDIR /app
COPY . .
RUN PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace && \
    pip install --no-cache-dir httpx uvicorn pyyaml
ENV SERVER_HOST=0.0.0.0 SERVER_PORT=8080 LOCAL_ENABLED=true
EXPOSE 8080 9464
ENTRYPOINT ["python", "-m", "inference_gateway_amd"]
