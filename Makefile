# Developer entry points (reference analog: the kubebuilder Makefile).

PYTHON ?= python
export PYTORCH_ROCM_ARCH ?= gfx950

.PHONY: build test test-gpu test-e2e bench bench-suite fmt clean \
        docker-build manifests docs deploy undeploy helm-install

build:  ## compile the native extension in-tree (hipcc cross-compiles gfx950)
	$(PYTHON) setup.py build_ext --inplace

test: build  ## CPU test tiers: unit + component + contract
	$(PYTHON) -m pytest tests/ -q -m "not gpu" --ignore=tests/test_e2e.py

test-e2e: build  ## in-process e2e (emulator + promlib + controller, ~60 s)
	$(PYTHON) -m pytest tests/test_e2e.py -q

test-gpu: build  ## GPU tiers (run on an MI355X box)
	$(PYTHON) -m pytest tests/ -q -m gpu

bench: build  ## headline benchmark (solver wall-clock + SLO attainment)
	$(PYTHON) bench.py --steps 20 --warmup 5

bench-suite: build  ## all five BASELINE configs
	$(PYTHON) tools/bench_suite.py

docker-build:  ## controller image (slim python; control plane is torch-free)
	docker build -t wva-amd-controller:latest .

manifests:  ## regenerate CRD schema docs from the API types (drift-tested)
	$(PYTHON) hack/gen_crd_docs.py

deploy:  ## apply CRD + controller + ConfigMaps to the current kube context
	kubectl apply -k deploy/

undeploy:
	kubectl delete -k deploy/ --ignore-not-found=true

helm-install:  ## install the chart (dev profile: see charts/.../values-dev.yaml)
	helm upgrade --install wva charts/workload-variant-autoscaler \
	  -n workload-variant-autoscaler-system --create-namespace

clean:
	rm -rf build wva_amd/*.so wva_amd/__pycache__ .pytest_cache
