# Convenience targets (see docs/API.md; the driver contract is bench.py /
# __graft_entry__.py, not this file).

.PHONY: build test test-gpu bench smoke soak fresh-check clean

build:
	python build_hip.py

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:          # on an MI355X box
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py --gpus 1 --steps 2000 --warmup 500

smoke:             # on an MI355X box
	python __graft_entry__.py smoke

soak:
	for i in 1 2 3; do python -m pytest tests/ -q -m "not gpu" --timeout 250 || exit 1; done

fresh-check:       # reproducibility: clone HEAD, build, run the CPU tier
	rm -rf /tmp/asyncframework_fresh
	git clone -q . /tmp/asyncframework_fresh
	cd /tmp/asyncframework_fresh && python -c "import __graft_entry__; __graft_entry__.build()" && python -m pytest tests/ -x -q -m "not gpu"
	rm -rf /tmp/asyncframework_fresh

clean:
	rm -rf asyncframework_amd/_dist_build asyncframework_amd/*.so build
