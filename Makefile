# Build the native core: the pybind11 extension (in-tree) and the standalone
# binaries (registrard daemon + zkensembled synthetic-ensemble server).
CXX ?= g++
CXXFLAGS ?= -std=c++17 -O2 -g -Wall -pthread
SRCDIR := registrar_amd/csrc
BINDIR := bin

CORE_SRCS := $(SRCDIR)/ensemble.cpp $(SRCDIR)/zkclient.cpp $(SRCDIR)/registrar.cpp \
             $(SRCDIR)/health.cpp $(SRCDIR)/orchestrator.cpp $(SRCDIR)/gpu.cpp
HDRS := $(wildcard $(SRCDIR)/*.hpp)

.PHONY: all ext daemon test check clean tsan asan stress

all: ext daemon

# the reference's `make check` (lint+style CI gate, Jenkinsfile:24-50)
# maps to the full local pipeline: build, CPU suite, sanitizers, bench smoke
check:
	bash tools/ci.sh

ext:
	python3 setup.py build_ext --inplace

daemon: $(BINDIR)/registrard $(BINDIR)/zkensembled

$(BINDIR)/registrard: $(SRCDIR)/daemon.cpp $(CORE_SRCS) $(HDRS)
	@mkdir -p $(BINDIR)
	$(CXX) $(CXXFLAGS) -o $@ $(SRCDIR)/daemon.cpp $(CORE_SRCS)

$(BINDIR)/zkensembled: $(SRCDIR)/ensemble_main.cpp $(CORE_SRCS) $(HDRS)
	@mkdir -p $(BINDIR)
	$(CXX) $(CXXFLAGS) -o $@ $(SRCDIR)/ensemble_main.cpp $(CORE_SRCS)

test:
	python3 -m pytest tests/ -x -q -m "not gpu"

# concurrency stress under sanitizers: the multi-threaded core
# (IO pool, sharded state, cross-thread watch delivery) under chaos
stress: $(BINDIR)/stress
	$(BINDIR)/stress -c 4 -t 8

$(BINDIR)/stress: $(SRCDIR)/stress_main.cpp $(CORE_SRCS) $(HDRS)
	@mkdir -p $(BINDIR)
	$(CXX) $(CXXFLAGS) -o $@ $(SRCDIR)/stress_main.cpp $(CORE_SRCS)

# gcc-11's libtsan mismodels this glibc's condition_variable timed waits
# (verified false positives on a textbook cv program), so sanitizer builds
# use ROCm's LLVM toolchain instead.
SANCXX ?= /opt/rocm/lib/llvm/bin/clang++

tsan: $(SRCDIR)/stress_main.cpp $(CORE_SRCS) $(HDRS)
	@mkdir -p $(BINDIR)
	$(SANCXX) -std=c++17 -O1 -g -fsanitize=thread -pthread -o $(BINDIR)/stress_tsan \
		$(SRCDIR)/stress_main.cpp $(CORE_SRCS)
	TSAN_OPTIONS="halt_on_error=1" $(BINDIR)/stress_tsan -c 4 -t 8

asan: $(SRCDIR)/stress_main.cpp $(CORE_SRCS) $(HDRS)
	@mkdir -p $(BINDIR)
	$(SANCXX) -std=c++17 -O1 -g -fsanitize=address,undefined -fno-sanitize-recover=all -pthread \
		-o $(BINDIR)/stress_asan $(SRCDIR)/stress_main.cpp $(CORE_SRCS)
	$(BINDIR)/stress_asan -c 4 -t 8

# release tarball (the reference's `make release`, Makefile:73-92 equivalent)
VERSION := $(shell python3 -c "import re;print(re.search(r'\"(.*)\"',open('registrar_amd/_version.py').read()).group(1))")
release: all
	@mkdir -p dist
	tar czf dist/mi355x-registrar-$(VERSION).tar.gz \
		bin/registrard bin/zkensembled etc/ docs/ README.md \
		registrar_amd/*.py registrar_amd/*.so registrar_amd/csrc \
		setup.py Makefile
	@echo "dist/mi355x-registrar-$(VERSION).tar.gz"

clean:
	rm -rf build bin dist registrar_amd/*.so
