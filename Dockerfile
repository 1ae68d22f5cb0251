# Builder: compile the native hot-path extension (_amcore) so the shipped
# image runs the same code path as the bench host — the reference ships one
# self-contained static binary (Dockerfile:21); the equivalent here is an
# image that never silently falls back to the pure-Python copy path.
FROM python:3.10-slim AS builder
RUN apt-get update && apt-get install -y --no-install-recommends gcc libc6-dev \
    && rm -rf /var/lib/apt/lists/*
WORKDIR /build
COPY pyproject.toml README.md setup.py ./
COPY native ./native
COPY active_monitor_amd ./active_monitor_amd
RUN pip install --no-cache-dir build wheel setuptools \
    && python setup.py build_ext --inplace \
    && test -f active_monitor_amd/_amcore*.so \
    && pip wheel --no-deps --no-build-isolation -w /wheels .

FROM python:3.10-slim
WORKDIR /app
COPY --from=builder /wheels /wheels
RUN pip install --no-cache-dir /wheels/*.whl && rm -rf /wheels
# fail loudly if the native extension ever goes missing (no silent fallback)
ENV AM_REQUIRE_NATIVE=1
# sanity check at build time: the wheel carries the compiled extension
RUN python -c "import active_monitor_amd._amcore"
USER 65532:65532
ENTRYPOINT ["active-monitor-amd"]
