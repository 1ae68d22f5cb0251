# Developer entry points (parity with the reference Makefile's test/build/run
# targets, adapted to the Python stack).
PY ?= python3

.PHONY: test test-gpu bench run manifests native docker-build sweep soak e2e-kind lint

test:
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests/ -q -m gpu

bench:
	$(PY) bench.py --steps 10 --warmup 2

sweep:
	$(PY) benchmarks/worker_sweep.py

run:
	$(PY) -m active_monitor_amd.cmd.main --backend memory --max-workers 10

native:
	$(PY) setup.py build_ext --inplace

manifests:
	$(PY) -m active_monitor_amd.api.crd > config/crd/bases/activemonitor.keikoproj.io_healthchecks.yaml

docker-build:
	docker build -t active-monitor-amd:latest .

soak:
	$(PY) benchmarks/soak.py --crs 1000 --repeat 5 --duration 300

e2e-kind:
	hack/e2e-kind.sh

lint:
	ruff check --select E9,F63,F7,F82 .
