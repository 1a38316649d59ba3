# lws-amd developer targets (reference Makefile analogue)
PY ?= python3

.PHONY: build test test-gpu bench smoke lint plan-steps manager clean

build:          ## build the gfx950 HIP kernel extension in-tree
	$(PY) -m lws_amd.ops.build

test:           ## CPU test tier (control plane, engine reference path, gloo TP)
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:       ## MI355X tier (kernel numerics vs fp32 reference, HIP engine)
	$(PY) -m pytest tests -q -m gpu

bench:          ## flagship benchmark (BASELINE.json metric), 1 GPU
	$(PY) bench.py --gpus 1 --steps 3 --warmup 1

smoke:
	$(PY) -c "import __graft_entry__ as g; g.build(); g.smoke()"

manager:        ## run the controller manager + API server
	$(PY) -m lws_amd --config examples/manager-config.yaml

plan-steps:     ## offline DS rollout previewer (hack/plan-steps analogue)
	$(PY) -m lws_amd.controllers.disaggregatedset.plan_steps \
	    --source '[2,6]' --target '[2,6]'

clean:
	rm -rf lws_amd/ops/build lws_amd/ops/_C.so .pytest_cache
