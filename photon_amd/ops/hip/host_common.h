// Host-side helpers shared by the .hip translation units.
#pragma once

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace photon_hip {

inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

#define DISPATCH_DTYPE(TENSOR, NAME, ...)                              \
  if ((TENSOR).scalar_type() == at::kBFloat16) {                       \
    using scalar_t = unsigned short;                                   \
    __VA_ARGS__;                                                       \
  } else if ((TENSOR).scalar_type() == at::kFloat) {                   \
    using scalar_t = float;                                            \
    __VA_ARGS__;                                                       \
  } else {                                                             \
    TORCH_CHECK(false, NAME ": unsupported dtype ", (TENSOR).dtype()); \
  }

}  // namespace photon_hip
