# Container image for the generator CLI
# (analog of the reference's alpine runner Dockerfile)
FROM python:3.10-slim

WORKDIR /app
COPY pyproject.toml README.md ./
COPY operator_builder_amd/ operator_builder_amd/
RUN pip install --no-cache-dir .

WORKDIR /workdir
ENTRYPOINT ["operator-builder"]
CMD ["--help"]
