# Build the native core: the pybind11 extension (in-tree) and the standalone
# binaries (registrard daemon + zkensembled synthetic-ensemble server).
CXX ?= g++
CXXFLAGS ?= -std=c++17 -O2 -g -Wall -pthread
SRCDIR := registrar_amd/csrc
BINDIR := bin

CORE_SRCS := $(SRCDIR)/ensemble.cpp $(SRCDIR)/zkclient.cpp $(SRCDIR)/registrar.cpp \
             $(SRCDIR)/health.cpp $(SRCDIR)/orchestrator.cpp $(SRCDIR)/gpu.cpp
HDRS := $(wildcard $(SRCDIR)/*.hpp)

.PHONY: all ext daemon test clean

all: ext daemon

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

clean:
	rm -rf build bin registrar_amd/*.so
