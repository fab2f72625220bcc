CXX ?= g++
CXXFLAGS ?= -O2 -std=c++17 -Wall -Wextra -g
COMMON := native/common/minijson.hpp native/common/util.hpp

.PHONY: all native test clean

all: native

native: native/bin/ckrt native/bin/ckd

native/bin/ckrt: native/ckrt/ckrt.cpp $(COMMON)
	@mkdir -p native/bin
	$(CXX) $(CXXFLAGS) -o $@ native/ckrt/ckrt.cpp

native/bin/ckd: native/ckd/ckd.cpp $(COMMON)
	@mkdir -p native/bin
	$(CXX) $(CXXFLAGS) -o $@ native/ckd/ckd.cpp

test: native
	python -m pytest tests/ -x -q -m "not gpu"

clean:
	rm -rf native/bin build
