PYTHON ?= python
IMG ?= inferno-amd/controller:latest

.PHONY: all build test test-gpu bench bench-cpu prof deploy-crd deploy undeploy docker-build lint

all: build test

## Build the HIP kernel library for gfx950 (in-tree .so)
build:
	$(PYTHON) -m inferno_amd.ops.build

## CPU test suite (no GPU required)
test:
	$(PYTHON) -m pytest tests/ -x -q -m "not gpu"

## CPU suite across 4 workers (developer loop; ~40% faster)
test-fast:
	$(PYTHON) -m pytest tests/ -q -m "not gpu" -n 4

## GPU differential + smoke tests (requires MI355X)
test-gpu:
	$(PYTHON) -m pytest tests/ -x -q -m gpu

## Full-loop out-of-process e2e: real controller process against the
## kube-apiserver stand-in + vLLM emulator + TLS Prometheus stand-in
## (envtest/Kind-equivalent tier; no cluster binaries needed)
test-e2e:
	$(PYTHON) -m pytest tests/test_e2e_apiserver.py tests/test_e2e_controller.py -x -q

## Flagship benchmark: 512-model fleet, GPU sweep
bench: build
	$(PYTHON) bench.py --steps 30 --warmup 5

## CPU golden baseline on the same fleet (slow)
bench-cpu:
	$(PYTHON) bench.py --backend cpu --models-per-gpu 64 --steps 3 --warmup 1

## rocprofv3 kernel profile of the bench (run on a GPU box)
prof:
	cd /tmp && TMPDIR=/tmp rocprofv3 --kernel-trace --stats -d $(CURDIR)/gpurun_out/prof \
		-- $(PYTHON) $(CURDIR)/bench.py --steps 5 --warmup 2

deploy-crd:
	kubectl apply -f deploy/crd/llmd.ai_variantautoscalings.yaml

deploy: deploy-crd
	kubectl apply -f deploy/controller.yaml
	kubectl apply -f deploy/configmap-accelerator-unitcost.yaml
	kubectl apply -f deploy/configmap-serviceclass.yaml

deploy-emulated: deploy
	kubectl apply -f deploy/emulator.yaml
	kubectl apply -f deploy/examples/vllme-variantautoscaling.yaml

undeploy:
	kubectl delete -f deploy/controller.yaml --ignore-not-found
	kubectl delete -f deploy/crd/llmd.ai_variantautoscalings.yaml --ignore-not-found

docker-build:
	docker build -t $(IMG) .

lint:
	$(PYTHON) -m compileall -q inferno_amd bench.py __graft_entry__.py

## AddressSanitizer pass over the native C++ greedy (host code)
asan-greedy:
	$(PYTHON) scripts/greedy_asan_check.py

## One-command local stack (apiserver stand-in + emulator + TLS prom + controller)
stack:
	bash deploy/install.sh --local

stack-smoke:
	bash deploy/install.sh --local --smoke
