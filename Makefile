# Convenience targets (the canonical entry points are __graft_entry__.build()
# and bench.py)
.PHONY: build test test-gpu bench probes clean

build:
	python -m cpd_amd.ops.build

test: build
	python -m pytest tests -q -m "not gpu"

test-gpu: build
	python -m pytest tests -q -m gpu

bench: build
	python bench.py

probes:
	hipcc --offload-arch=gfx950 -O3 -std=c++17 -Wno-unused-value tools/gemm_probe.hip -o tools/gemm_probe
	hipcc --offload-arch=gfx950 -O3 -std=c++17 -Wno-unused-value tools/quant_gemm_probe.hip -o tools/quant_gemm_probe
	hipcc --offload-arch=gfx950 -O3 -std=c++17 -Wno-unused-value tools/elem_probe.hip -o tools/elem_probe

clean:
	rm -f cpd_amd/ops/_cpd_cpu.so cpd_amd/ops/_cpd_hip.so tools/gemm_probe tools/quant_gemm_probe tools/elem_probe
