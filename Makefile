.PHONY: ext test test-gpu bench serve-bench lint docker

ext:
	PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py --steps 2 --warmup 1

serve-bench:
	python scripts/bench_serving.py --rounds 200 --threads 8 --endpoint both

lint:
	python -m pytest tests/test_formatting.py -q

sanitize:
	bash scripts/gpu_sanitize.sh

docker:
	docker build -t gordo-amd .
