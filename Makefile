# creditcore — common entry points (see docs/runbook.md)

lint:
	python tools/lint.py

.PHONY: lint build test test-gpu bench serve train smoke pipeline clean

build:
	python setup.py build_ext --inplace

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py --gpus 1 --steps 400 --warmup 50

train:
	python -m creditcore train --model-dir ./model --max-evals 10

serve:
	python -m creditcore serve --model-directory ./model --port 5000

smoke:
	python -m creditcore smoke --url http://127.0.0.1:5000

pipeline:
	python -m creditcore pipeline --auto-approve

clean:
	rm -rf build creditcore/_ccore*.so .pytest_cache
