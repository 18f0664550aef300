# Developer entry points (reference Makefile: build/test/codegen/manifests).
PYTHON ?= python3

.PHONY: test test-fast test-race bench bench-full soak e2e run-controller run-webhook manifests-validate

test:
	$(PYTHON) -m pytest tests/ -q -m "not gpu"

test-fast:
	$(PYTHON) -m pytest tests/ -q -m "not gpu" -n 4

# threading stress profile: repeat the concurrency-heavy suites
test-race:
	# lock-order checking (agac/lockcheck.py): every project lock is
	# instrumented; a lock-order cycle (potential deadlock) fails the
	# session (the go-test-race analogue this pure-Python tier can have).
	# One full-suite pass + 2x stress repetition for flake surfacing.
	AGAC_LOCKCHECK=1 $(PYTHON) -m pytest tests/ -q -m "not gpu"
	for i in 1 2; do \
		AGAC_LOCKCHECK=1 $(PYTHON) -m pytest tests/test_stress_concurrency.py \
			tests/test_chaos_recovery.py tests/test_store_event_sourcing.py \
			tests/test_fault_injection.py tests/test_cloud_resync.py \
			tests/test_failover_under_churn.py -q || exit 1; \
	done

bench:
	$(PYTHON) bench.py --steps 10 --warmup 3

bench-full:
	$(PYTHON) bench.py --steps 10 --warmup 3 --scenario full

soak:
	$(PYTHON) hack/soak.py --minutes 2 --objects 64 --scenario full

e2e:
	bash hack/run-e2e.sh

run-controller:
	$(PYTHON) -m agac.cli -v controller --api memory --no-leader-elect

run-webhook:
	$(PYTHON) -m agac.cli -v webhook --no-ssl --port 8443

manifests-validate:
	$(PYTHON) -c "import yaml,glob; [list(yaml.safe_load_all(open(f))) for f in glob.glob('config/**/*.yaml', recursive=True)]; print('manifests OK')"
	$(PYTHON) hack/helm_render.py --validate
