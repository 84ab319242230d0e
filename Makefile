# binder-amd: native-first rebuild of TritonDataCenter/binder.
# Everything is plain C++20 + POSIX; no external libraries.

CXX ?= g++
PYTHON ?= python3
CXXFLAGS ?= -O2 -g -std=c++20 -fPIC -fno-omit-frame-pointer \
	-Wall -Wextra -Werror -Wno-unused-parameter -MMD -MP
LDFLAGS ?=

BUILD := build

CORE_SRCS := \
	native/common/json.cpp \
	native/common/log.cpp \
	native/common/loop.cpp \
	native/dns/codec.cpp \
	native/engine/store.cpp \
	native/engine/engine.cpp

SERVER_SRCS := \
	native/server/metrics.cpp \
	native/server/server.cpp \
	native/server/recursion.cpp \
	native/server/ldap.cpp \
	native/zk/client.cpp \
	native/zk/mirror.cpp

CORE_OBJS := $(CORE_SRCS:%.cpp=$(BUILD)/%.o)
SERVER_OBJS := $(SERVER_SRCS:%.cpp=$(BUILD)/%.o)

PY_EXT_SUFFIX := $(shell $(PYTHON)-config --extension-suffix 2>/dev/null || echo .so)
PY_INCLUDES := $(shell $(PYTHON) -m pybind11 --includes)
PYMOD := binder_amd/_native$(PY_EXT_SUFFIX)

BINARIES := bin/binderd bin/binder-balancer bin/dnsblast \
	bin/binder-adjust bin/binder-supervisor bin/zklogcat bin/zktool \
	bin/zkd

all: $(PYMOD) $(BINARIES)

bin/binderd: $(CORE_OBJS) $(SERVER_OBJS) $(BUILD)/native/server/binderd_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS) -lssl -lcrypto -lpthread

bin/binder-balancer: $(CORE_OBJS) $(BUILD)/native/balancer/balancer_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

bin/dnsblast: $(CORE_OBJS) $(BUILD)/native/bench/dnsblast_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS) -lpthread

bin/binder-adjust: $(CORE_OBJS) $(BUILD)/native/adjust/adjust_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

bin/binder-supervisor: $(CORE_OBJS) $(BUILD)/native/adjust/supervisor_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

bin/zklogcat: $(CORE_OBJS) $(BUILD)/native/zklog/zklogcat_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

bin/zktool: $(CORE_OBJS) $(BUILD)/native/zk/client.o $(BUILD)/native/zk/zktool_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

bin/zkd: $(CORE_OBJS) $(BUILD)/native/zkd/zkd_main.o
	@mkdir -p bin
	$(CXX) $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

$(BUILD)/%.o: %.cpp
	@mkdir -p $(dir $@)
	$(CXX) $(CXXFLAGS) -c $< -o $@

# pybind11 module (compiled separately: needs Python includes, and
# -Wno-error for pybind's warnings under -Wextra)
$(BUILD)/native/pybind/module.o: native/pybind/module.cpp
	@mkdir -p $(dir $@)
	$(CXX) $(CXXFLAGS) -Wno-error $(PY_INCLUDES) -c $< -o $@

$(PYMOD): $(CORE_OBJS) $(BUILD)/native/pybind/module.o
	$(CXX) -shared $(CXXFLAGS) $^ -o $@ $(LDFLAGS)

# `make test` parity with the reference's nodeunit target
# (Makefile:169-171 there)
test: all
	$(PYTHON) -m pytest tests -q -m "not gpu"

# Lint/style gate (parity with the reference's eslint/jsstyle/cstyle
# gates, /root/reference/Makefile:17-20): C++ and Python style checks in
# tools/lint.py; -Werror already covers compiler diagnostics.
check: all
	$(PYTHON) -m py_compile binder_amd/*.py bench.py __graft_entry__.py
	$(PYTHON) tools/lint.py
	@echo "check OK"

# release tarball layout under /opt/binder-amd (the reference ships
# /opt/smartdc/binder with balancer+smf_adjust in lib/ and zklog in
# bin/; Makefile:178-229 there)
release: all
	rm -rf dist/binder-amd
	mkdir -p dist/binder-amd/bin dist/binder-amd/etc
	cp bin/binderd bin/binder-balancer bin/binder-adjust \
	    bin/zkd \
	    bin/binder-supervisor bin/zklogcat bin/dnsblast bin/zktool \
	    dist/binder-amd/bin/
	cp -r deploy dist/binder-amd/
	cp -r tools dist/binder-amd/
	cp etc/config.json.in etc/config.sample.json dist/binder-amd/etc/
	mkdir -p dist/binder-amd/lib/python/binder_amd
	cp binder_amd/*.py $(PYMOD) dist/binder-amd/lib/python/binder_amd/
	cp README.md LICENSE dist/binder-amd/
	cp -r docs dist/binder-amd/
	tar -C dist -czf dist/binder-amd.tar.gz binder-amd
	@echo "release: dist/binder-amd.tar.gz"

clean:
	rm -rf $(BUILD) $(PYMOD) $(BINARIES) dist

-include $(shell find $(BUILD) -name '*.d' 2>/dev/null)

.PHONY: all clean test check release
