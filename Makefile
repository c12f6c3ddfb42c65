# BNG-AMD Makefile (the reference ships the same operator entry
# points: build / test / demo / clean — Makefile:1-60)

PYTHON ?= python3
ARCH   ?= gfx950

.PHONY: help build test test-gpu bench demo lint clean

help:
	@echo "BNG-AMD (MI355X-native)"
	@echo ""
	@echo "  make build      - compile the HIP dataplane extension ($(ARCH))"
	@echo "  make test       - CPU test suite (no GPU needed)"
	@echo "  make test-gpu   - GPU test suite (needs an MI355X)"
	@echo "  make bench      - single-GPU benchmark (driver contract)"
	@echo "  make demo       - subscriber-lifecycle demo (no dataplane)"
	@echo "  make clean      - remove build artifacts"

build:
	PYTORCH_ROCM_ARCH=$(ARCH) $(PYTHON) -c "import __graft_entry__ as g; g.build()"

test:
	$(PYTHON) -m pytest tests/ -x -q -m "not gpu"

test-gpu:
	$(PYTHON) -m pytest tests/ -x -q -m gpu

bench:
	$(PYTHON) bench.py --gpus 1

demo:
	$(PYTHON) -m bng_amd.cli.main demo --subscribers 10

clean:
	rm -rf bng_amd/dataplane/csrc/*.o bng_amd/dataplane/*.so \
	       bng_amd/dataplane/csrc/*.so build/ dist/ \
	       $$(find . -name __pycache__ -type d)
