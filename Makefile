# Test/bench slices (reference: Makefile targets test_core/test_fsdp/...)

.PHONY: quality test test_core test_distributed test_big_modeling test_kernels test_gpu bench build

build:
	python setup.py build_ext --inplace

test:
	python -m pytest tests/ -q -m "not gpu"

test_core:
	python -m pytest tests/test_accelerator.py tests/test_state.py tests/test_data_loader.py \
	  tests/test_operations.py tests/test_misc.py tests/test_tracking.py -q

test_distributed:
	python -m pytest tests/test_grad_sync.py tests/test_fsdp.py tests/test_ep.py -q -m "not gpu"

test_big_modeling:
	python -m pytest tests/test_big_modeling.py tests/test_hooks.py tests/test_modeling_utils.py \
	  tests/test_quantization.py -q -m "not gpu"

test_kernels:
	python -m pytest tests/test_kernels.py tests/test_norms.py tests/test_attention.py \
	  tests/test_fp8.py tests/test_mfma_gemm.py -q -m "not gpu"

test_gpu:
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py
