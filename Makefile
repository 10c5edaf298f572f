PY ?= python3

.PHONY: build test test-gpu bench serve worker clean

build:
	$(PY) -m code_intelligence_amd.ops.build

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --steps 10 --warmup 3

serve:
	$(PY) -m code_intelligence_amd.serve.app --model_path $(MODEL_PATH)

worker:
	$(PY) -m code_intelligence_amd.label.worker

clean:
	rm -rf code_intelligence_amd/ops/_build code_intelligence_amd/ops/*.so

# repeated-suite stability soak (flake hunting)
soak:
	for i in 1 2 3; do $(PY) -m pytest tests -q -m "not gpu" -p no:cacheprovider | tail -1; done

# per-kernel microbenchmarks (GPU box)
kernel-bench:
	$(PY) scripts/kernel_bench.py
