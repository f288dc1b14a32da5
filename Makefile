# kuberay-amd developer entry points

.PHONY: build test test-gpu bench crds soak lint

build:            ## build the C++/HIP extensions in-tree (gfx950)
	python -m kuberay_amd._native.build

test:             ## CPU test suite (the driver's per-round check)
	python -m pytest tests -x -q -m "not gpu"

test-gpu:         ## on an MI355X box only
	python -m pytest tests -x -q -m gpu

bench:            ## driver-contract benchmark, default flags
	python bench.py

soak:             ## 5-minute 500-cluster churn soak
	python benchmark/perf-tests/soak.py --minutes 5 --clusters 500

crds:             ## regenerate deploy/crds from the pydantic models
	python -m kuberay_amd.crds deploy/crds

sharded:          ## sharded-operator topology bench over the HTTP facade
	python benchmark/perf-tests/sharded.py --shards 2 --clusters 200

upgrade-storm:    ## 100 simultaneous zero-downtime RayService upgrades
	python benchmark/perf-tests/rayservice_upgrade.py --services 100
