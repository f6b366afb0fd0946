PYTHON ?= python3
IMAGE  ?= ghcr.io/example/kata-xpu-device-plugin-amd:0.1.0

.PHONY: build build-hip test test-gpu bench soak image clean doctor validate burnin partition

build:
	$(PYTHON) setup.py build_ext --inplace

build-hip:
	$(PYTHON) -c "from setup import build_hip; print(build_hip('gfx950'))"

test:
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests/ -q -m gpu

bench:
	$(PYTHON) bench.py --steps 30 --warmup 5

doctor:
	$(PYTHON) -m kata_xpu_device_plugin_amd.tools.doctor

validate:
	$(PYTHON) -m kata_xpu_device_plugin_amd.tools.validate

burnin:
	$(PYTHON) -m kata_xpu_device_plugin_amd.tools.burnin

partition:
	$(PYTHON) -m kata_xpu_device_plugin_amd.tools.partition show

soak:
	$(PYTHON) benchmarks/churn_soak.py --minutes 10 --clients 3

image:
	docker build -t $(IMAGE) .

clean:
	rm -rf build kata_xpu_device_plugin_amd/*.so kata_xpu_device_plugin_amd/**/__pycache__
