build-asan/test_ddp_fake: tests/cpp/test_ddp_fake.cc src/comm/comm.h \
  src/core/tensor.h src/core/common.h src/models/models.h \
  src/nn/modules.h src/autograd/tape.h src/core/ops.h \
  src/train/trainer.h src/nn/optim.h src/train/ddp.h src/util/util.h
src/comm/comm.h:
src/core/tensor.h:
src/core/common.h:
src/models/models.h:
src/nn/modules.h:
src/autograd/tape.h:
src/core/ops.h:
src/train/trainer.h:
src/nn/optim.h:
src/train/ddp.h:
src/util/util.h:
