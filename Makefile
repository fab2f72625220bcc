CXX ?= g++
CXXFLAGS ?= -O2 -std=c++17 -Wall -Wextra -g
COMMON := native/common/minijson.hpp native/common/util.hpp

PY ?= python3
EXT_SUFFIX := $(shell $(PY) -c "import sysconfig; print(sysconfig.get_config_var('EXT_SUFFIX'))")
PYBIND_INC := $(shell $(PY) -c "import pybind11; print(pybind11.get_include())")
PY_INC := $(shell $(PY) -c "import sysconfig; print(sysconfig.get_paths()['include'])")

.PHONY: all native pymod test test-gpu bench soak docs clean

all: native pymod

native: native/bin/ckrt native/bin/ckd native/bin/ckgw native/bin/devbpf_probe

native/bin/ckgw: native/ckgw/ckgw.cpp $(COMMON)
	@mkdir -p native/bin
	$(CXX) $(CXXFLAGS) -o $@ native/ckgw/ckgw.cpp

pymod: clawker_amd/_native$(EXT_SUFFIX)

clawker_amd/_native$(EXT_SUFFIX): native/pymod/native.cpp
	$(CXX) $(CXXFLAGS) -shared -fPIC -I$(PYBIND_INC) -I$(PY_INC) \
		-o $@ native/pymod/native.cpp

native/bin/ckrt: native/ckrt/ckrt.cpp native/ckrt/devbpf.hpp $(COMMON)
	@mkdir -p native/bin
	$(CXX) $(CXXFLAGS) -o $@ native/ckrt/ckrt.cpp

native/bin/ckd: native/ckd/ckd.cpp $(COMMON)
	@mkdir -p native/bin
	$(CXX) $(CXXFLAGS) -o $@ native/ckd/ckd.cpp

native/bin/devbpf_probe: native/tests/devbpf_probe.cpp native/ckrt/devbpf.hpp
	@mkdir -p native/bin
	$(CXX) $(CXXFLAGS) -o $@ native/tests/devbpf_probe.cpp

test: native pymod
	python -m pytest tests/ -x -q -m "not gpu"

# ---- sanitizer lane: ASAN+UBSAN builds of the native
# runtime + minijson fuzzer, exercised by the robustness suite ----------
ASAN_FLAGS := -O1 -std=c++17 -Wall -Wextra -g -fsanitize=address,undefined \
	-fno-sanitize-recover=all -fno-omit-frame-pointer
ASAN_BIN := native/bin-asan

native-asan: $(ASAN_BIN)/ckrt $(ASAN_BIN)/ckd $(ASAN_BIN)/ckgw $(ASAN_BIN)/fuzz_minijson

$(ASAN_BIN)/ckrt: native/ckrt/ckrt.cpp native/ckrt/devbpf.hpp $(COMMON)
	@mkdir -p $(ASAN_BIN)
	$(CXX) $(ASAN_FLAGS) -o $@ native/ckrt/ckrt.cpp

$(ASAN_BIN)/ckd: native/ckd/ckd.cpp $(COMMON)
	@mkdir -p $(ASAN_BIN)
	$(CXX) $(ASAN_FLAGS) -o $@ native/ckd/ckd.cpp

$(ASAN_BIN)/ckgw: native/ckgw/ckgw.cpp $(COMMON)
	@mkdir -p $(ASAN_BIN)
	$(CXX) $(ASAN_FLAGS) -o $@ native/ckgw/ckgw.cpp

$(ASAN_BIN)/fuzz_minijson: native/tests/fuzz_minijson.cpp $(COMMON)
	@mkdir -p $(ASAN_BIN)
	$(CXX) $(ASAN_FLAGS) -o $@ native/tests/fuzz_minijson.cpp

test-asan: native-asan
	$(ASAN_BIN)/fuzz_minijson 200000
	ASAN_OPTIONS=detect_leaks=0 CLAWKER_NATIVE_BIN=$(abspath $(ASAN_BIN)) \
		python -m pytest tests/test_robustness.py tests/test_proc_backend.py \
		tests/test_unprivileged_agent.py tests/test_firewall_e2e.py \
		tests/test_ssh_egress.py -x -q -m "not gpu"

test-gpu: native pymod
	python -m pytest tests/ -x -q -m gpu

bench: native pymod
	python bench.py

soak: native pymod
	python tools/soak.py 1000 /tmp/clawker-soak.json

docs:
	python tools/gen_docs.py

clean:
	rm -rf native/bin build
