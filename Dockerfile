FROM python:3.10-slim
WORKDIR /app
COPY pyproject.toml README.md ./
COPY active_monitor_amd ./active_monitor_amd
RUN pip install --no-cache-dir .
USER 65532:65532
ENTRYPOINT ["active-monitor-amd"]
