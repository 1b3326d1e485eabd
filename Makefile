# Convenience targets (the driver uses __graft_entry__.build() and the
# pytest/bench contracts directly; this Makefile just wraps them).
.PHONY: build test test-gpu bench clean

build:
	python -c "import __graft_entry__ as g; g.build()"

test:
	python -m pytest tests -q -m "not gpu"

test-gpu:
	python -m pytest tests -q -m gpu

bench:
	python bench.py

clean:
	rm -f neutronstarlite_amd/libnts_hip.so oracle/liboracle.so
	rm -rf cpp/build
