build-asan/test_core: tests/cpp/test_core.cc src/ckpt/npz.h \
  src/core/common.h src/core/ops.h src/core/tensor.h src/util/util.h
src/ckpt/npz.h:
src/core/common.h:
src/core/ops.h:
src/core/tensor.h:
src/util/util.h:
