# Convenience targets (see docs/general/installation_guide.md)
PY ?= python

.PHONY: build test test-gpu bench smoke parity bench-attn bench-decode family-smoke clean

build:
	PYTORCH_ROCM_ARCH=gfx950 $(PY) setup.py build_ext --inplace

test:
	$(PY) -m pytest tests -q -m "not gpu"

test-gpu:
	$(PY) -m pytest tests -q -m gpu

bench:
	$(PY) bench.py --gpus 1 --steps 8 --warmup 3

smoke:
	$(PY) __graft_entry__.py smoke

parity:
	$(PY) tools/check_grad_parity.py --layout tp2_sp
	$(PY) tools/check_grad_parity.py --layout lora_tp2_sp

bench-attn:
	$(PY) tools/bench_attn_kernels.py --iters 10

bench-decode:
	$(PY) tools/bench_decode.py

family-smoke:
	$(PY) tools/gpu_family_smoke.py

clean:
	rm -rf build neuronx_distributed_training_amd/ops/csrc/*.o
