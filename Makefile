# Developer entry points (the driver uses __graft_entry__.py + pytest
# directly; these are for humans).

.PHONY: build test test-gpu bench ci sanitize resources clean

build:
	PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py --gpus 1 --steps 8 --warmup 3

ci:
	bash scripts/ci.sh

sanitize:
	bash csrc/tools/sanitize_build.sh

resources:
	@for f in csrc/*.hip; do \
	  case $$f in *_hip.hip) continue;; esac; \
	  echo "== $$f"; bash csrc/tools/check_resources.sh $$f 2>&1 | \
	    grep -E "kernel-resource-usage|SGPRs|VGPRs|LDS|Occupancy" | head -20; \
	done

clean:
	rm -rf build vit_10b_fsdp_example_amd/_C*.so csrc/*_hip.hip
