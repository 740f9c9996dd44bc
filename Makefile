# Build/test entry points (the reference drives everything through its
# Makefile too). `make build` cross-compiles the gfx950 extension on any box.
PY ?= python

.PHONY: build test test-gpu bench clean install uninstall deploy undeploy docker-build run

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) -m dgl_operator_amd.csrc.build

test: build
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests -q -m gpu

bench: build
	$(PY) bench.py --gpus 1 --steps 30 --warmup 10

clean:
	rm -rf dgl_operator_amd/_C.so dgl_operator_amd/csrc/.obj

# cluster lifecycle (reference Makefile install/uninstall/deploy/undeploy)
install:
	kubectl apply -f deploy/crd/dgljobs.qihoo.net.yaml

uninstall:
	kubectl delete -f deploy/crd/dgljobs.qihoo.net.yaml --ignore-not-found

deploy: install
	kubectl apply -f deploy/v1alpha1/dgl-operator.yaml

undeploy:
	kubectl delete -f deploy/v1alpha1/dgl-operator.yaml --ignore-not-found

docker-build:
	docker build -f deploy/docker/Dockerfile.manager -t dgl-operator-amd/manager:latest .
	docker build -f deploy/docker/Dockerfile.watcher-loop -t dgl-operator-amd/watcher-loop:latest .
	docker build -f deploy/docker/Dockerfile.kubectl-download -t dgl-operator-amd/kubectl-download:latest .
	docker build -f deploy/docker/Dockerfile.worker -t dgl-operator-amd/worker:latest .

run:
	$(PY) -m dgl_operator_amd.operator_plane.manager --reconcile-interval 1.0
