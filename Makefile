# Convenience targets (the reference drives everything through makefiles;
# here they wrap the python entry points).
.PHONY: all build test test-gpu bench repro clean

all: build

build:
	python build.py

test: build
	python -m pytest tests -q -m "not gpu"

test-gpu: build
	python -m pytest tests -q -m gpu

bench: build
	python bench.py --gpus 1 --steps 25 --warmup 3

repro: build
	bash scripts/reproduce.sh quick

clean:
	rm -f gats_amd/_core*.so
	rm -rf build/obj
