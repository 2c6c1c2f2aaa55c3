# agent-bom MI355X runtime image.
# Base must provide ROCm >= 7.0 + PyTorch-ROCm (the gfx950 engine is
# compiled in-tree; hipcc cross-compiles without a GPU present).
ARG BASE=rocm/pytorch:latest
FROM ${BASE}

WORKDIR /opt/agent-bom
COPY pyproject.toml setup.py ./
COPY agentbom_amd ./agentbom_amd
COPY sdks ./sdks
COPY bench.py __graft_entry__.py ./

RUN pip install --no-build-isolation --no-deps -e . \
    && python -m agentbom_amd.ops.build || echo "hipcc unavailable: CPU-only image"

EXPOSE 8000
ENTRYPOINT ["agent-bom"]
CMD ["serve", "--host", "0.0.0.0", "--port", "8000"]
