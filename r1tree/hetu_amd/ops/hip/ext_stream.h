// Current-stream helper for ROCm torch.
#pragma once
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
inline hipStream_t hetu_current_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}
