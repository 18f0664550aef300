# Two-stage image (reference Dockerfile: golang builder -> distroless).
# Python equivalent: build a wheel, install into a slim runtime.
FROM python:3.10-slim AS builder
WORKDIR /src
COPY pyproject.toml ./
COPY agac ./agac
RUN pip install --no-cache-dir build && python -m build --wheel --outdir /dist

FROM python:3.10-slim
RUN useradd --uid 65532 --no-create-home nonroot
COPY --from=builder /dist/*.whl /tmp/
RUN pip install --no-cache-dir /tmp/*.whl && rm /tmp/*.whl
USER 65532:65532
ENTRYPOINT ["agac"]
CMD ["controller"]
