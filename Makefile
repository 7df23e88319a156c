# Developer targets (reference Makefile:1-6 role)
.PHONY: style test test-gpu build

style:
	python -m black zero_transformer_amd tests torch_compatability *.py || true

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

build:
	python -m zero_transformer_amd.ops.build
