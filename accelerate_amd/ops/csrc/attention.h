// Shared between attention_kernels.hip (kernels) and bindings.hip (host).
#pragma once
#include <cstdint>

// per-tensor element strides over (batch, head, seq); innermost dim is
// contiguous. Lets the flash-attention kernels read q/k/v views DIRECTLY
// (no .contiguous() copies, GQA kv head = h / (Hq/Hkv) without
// repeat_interleave) and write output in BSHD storage.
struct Str3 {
  int64_t b, h, s;
};
