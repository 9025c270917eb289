# Developer entry points for the generator itself
# (analog of the reference's Makefile: build/test/func-test targets)

PYTHON ?= python3

.PHONY: all test test-fast test-unit test-func test-gpu bench smoke lint parity release clean

all: lint test

test:
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

# parallel run (pytest-xdist); the suite is worker-safe
test-fast:
	$(PYTHON) -m pytest tests/ -q -m "not gpu" -n auto

# unit tiers only (marker engine, yamlast, domain model)
test-unit:
	$(PYTHON) -m pytest tests/test_lexer.py tests/test_marker_parser.py \
		tests/test_yamlast.py tests/test_api_fields.py tests/test_rbac.py \
		tests/test_companion_and_config.py tests/test_utils_and_misc.py -q

# functional tiers (full init + create api over fixtures)
test-func:
	$(PYTHON) -m pytest tests/test_generate_standalone.py \
		tests/test_generate_collection.py tests/test_generate_edge.py \
		tests/test_reference_fixtures.py tests/test_api_upgrade.py -q

test-gpu:
	$(PYTHON) -m pytest tests/ -q -m gpu

bench:
	$(PYTHON) bench.py --steps 20 --warmup 3

smoke:
	$(PYTHON) __graft_entry__.py

# AST-based lint gate (analog of the reference's golangci-lint step)
lint:
	$(PYTHON) scripts/lint.py

# render the reference templates against a generated tree (diff==0 check)
parity:
	$(PYTHON) -m pytest tests/test_reference_parity.py -q

# build distributables + shell completions (analog of .goreleaser.yml)
release:
	bash scripts/release.sh

clean:
	find . -name __pycache__ -type d -prune -exec rm -rf {} +
	rm -rf .pytest_cache build dist *.egg-info
