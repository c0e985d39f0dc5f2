# Developer entry points (gfx950 cross-compile works without a GPU).

.PHONY: build test test-gpu bench smoke lint clean

build:
	python tepdist_amd/ops/build_ext.py

test:
	python -m pytest tests/ -q -m "not gpu"

test-gpu:
	python -m pytest tests/ -q -m gpu

bench:
	python bench.py

smoke:
	python -c "import __graft_entry__ as g; g.build(); g.smoke()"

clean:
	rm -rf tepdist_amd/ops/csrc/build tepdist_amd/ops/_tepdist_hip.so \
	    tepdist_amd/runtime/_tepdist_rt.so
