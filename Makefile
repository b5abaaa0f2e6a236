# Convenience targets (the reference's paper/kernel/gpu/Makefile analog).
# All builds are in-tree; nothing is installed into site-packages.

PY ?= python

.PHONY: build test test-gpu selftest benchmark sweep profile profile-pmc clean

build:
	$(PY) -m gpudpf._build

test: build
	$(PY) -m pytest tests/ -q -m "not gpu"

test-gpu: build
	$(PY) -m pytest tests/ -q -m gpu

selftest: build
	$(PY) -m gpudpf.selftest

benchmark: build
	$(PY) benchmark.py

sweep: build
	bash benchmarks/sweep.sh AES128 fused

# rocprofv3 profiling (run on a GPU box; TMPDIR avoids rocprof tmp issues)
PROF_CMD ?= $(PY) bench.py --steps 3 --warmup 1 --entries 1048576 --prf AES128
profile: build
	TMPDIR=/tmp rocprofv3 --kernel-trace --stats -d profiles/rocprof -o run -- $(PROF_CMD)

profile-pmc: build
	TMPDIR=/tmp rocprofv3 --pmc SQ_INSTS_VALU,SQ_INSTS_LDS,SQ_LDS_BANK_CONFLICT,SQ_WAIT_ANY,SQ_WAVE_CYCLES,SQ_BUSY_CU_CYCLES -d profiles/rocprof -o pmc -- $(PROF_CMD)

clean:
	rm -f gpudpf/_core*.so gpudpf/_hip*.so

# host-side address sanitizer build of the CPU core (race/UB tier)
asan:
	g++ -O1 -g -fsanitize=address -std=c++17 -shared -fPIC -pthread \
	  -I$$(python -c "import pybind11;print(pybind11.get_include())") \
	  -I$$(python -c "import sysconfig;print(sysconfig.get_paths()['include'])") \
	  csrc/core/dpf_core.cc csrc/core/prf.cc csrc/core/aes128.cc \
	  csrc/core/core_bindings.cc -o /tmp/_core_asan.so
	@echo "ASan build OK (load with LD_PRELOAD=libasan.so python ...)"
