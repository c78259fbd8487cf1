# modelx_amd build
# CPU binaries (modelxd, modelx-s3d) build with g++; the GPU extension
# (_core) builds with hipcc via setup_ext.py (PYTORCH_ROCM_ARCH=gfx950).

CXX ?= g++
CXXFLAGS ?= -O2 -std=c++17 -Wall -Wno-unused-parameter -pthread
INCLUDES = -Icore/include
LIBS = -lcrypto -lssl -lz

BIN = bin
CORE_SRC = core/src/json.cpp core/src/http.cpp core/src/wire.cpp \
           core/src/store_local.cpp core/src/store_fs.cpp core/src/s3.cpp \
           core/src/sigv4.cpp core/src/auth.cpp core/src/registry.cpp

.PHONY: all clean servers ext test stamp servers-asan servers-tsan

# sanitizer builds of the C++ servers (SURVEY.md §5 race detection):
# tests/test_sanitizers.py runs the integration flow against these.
servers-asan: $(BIN)/modelxd-asan $(BIN)/modelx-s3d-asan
servers-tsan: $(BIN)/modelxd-tsan $(BIN)/modelx-s3d-tsan

$(BIN)/modelxd-asan: $(CORE_SRC) core/src/modelxd_main.cpp core/include/modelx/*.hpp
	@mkdir -p $(BIN)
	$(CXX) $(CXXFLAGS) -g -fsanitize=address $(INCLUDES) $(CORE_SRC) core/src/modelxd_main.cpp -o $@ $(LIBS)

$(BIN)/modelx-s3d-asan: $(CORE_SRC) core/src/s3d.cpp core/include/modelx/*.hpp
	@mkdir -p $(BIN)
	$(CXX) $(CXXFLAGS) -g -fsanitize=address $(INCLUDES) $(CORE_SRC) core/src/s3d.cpp -o $@ $(LIBS)

$(BIN)/modelxd-tsan: $(CORE_SRC) core/src/modelxd_main.cpp core/include/modelx/*.hpp
	@mkdir -p $(BIN)
	$(CXX) $(CXXFLAGS) -g -fsanitize=thread $(INCLUDES) $(CORE_SRC) core/src/modelxd_main.cpp -o $@ $(LIBS)

$(BIN)/modelx-s3d-tsan: $(CORE_SRC) core/src/s3d.cpp core/include/modelx/*.hpp
	@mkdir -p $(BIN)
	$(CXX) $(CXXFLAGS) -g -fsanitize=thread $(INCLUDES) $(CORE_SRC) core/src/s3d.cpp -o $@ $(LIBS)

all: servers ext

servers: $(BIN)/modelxd $(BIN)/modelx-s3d

$(BIN)/modelxd: $(CORE_SRC) core/src/modelxd_main.cpp core/include/modelx/*.hpp
	@mkdir -p $(BIN)
	$(CXX) $(CXXFLAGS) $(INCLUDES) $(CORE_SRC) core/src/modelxd_main.cpp -o $@ $(LIBS)

$(BIN)/modelx-s3d: $(CORE_SRC) core/src/s3d.cpp core/include/modelx/*.hpp
	@mkdir -p $(BIN)
	$(CXX) $(CXXFLAGS) $(INCLUDES) $(CORE_SRC) core/src/s3d.cpp -o $@ $(LIBS)

ext:
	python setup_ext.py build_ext --inplace

test:
	python -m pytest tests/ -x -q -m "not gpu"

stamp:
	@printf 'GIT_VERSION = "%s"\nGIT_COMMIT = "%s"\nBUILD_DATE = "%s"\n' \
	  "$$(git describe --tags --always 2>/dev/null || echo v0.1.0)" \
	  "$$(git rev-parse HEAD 2>/dev/null || echo unknown)" \
	  "$$(date -u +%Y-%m-%dT%H:%M:%SZ)" > modelx_amd/_build_stamp.py

clean:
	rm -rf $(BIN) build modelx_amd/_core*.so
