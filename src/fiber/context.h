// brpc_amd: fiber context primitives (see context.S).
#pragma once

#include <stddef.h>
#include <stdint.h>

extern "C" {
// Defined in context.S.
void* bam_jump_context(void** from_sp, void* to_sp, void* arg);
}

namespace bam {

// Builds an initial context on [stack_base, stack_base+size) that enters
// fn(arg_of_first_jump) when first jumped to. fn must never return.
inline void* make_context(void* stack_base, size_t size, void (*fn)(void*)) {
  uintptr_t top = ((uintptr_t)stack_base + size) & ~(uintptr_t)15;
  // Frame consumed by the restore side of bam_jump_context: 8 bytes fpu
  // state + 6 saved GPRs + return address = 64 bytes. Choose sp ≡ 8 mod 16
  // so the entry function sees a post-call-aligned stack.
  uint64_t* sp = (uint64_t*)(top - 72);
  sp[0] = 0x0000037F00001F80ULL;  // mxcsr (low 4B) = 0x1F80, x87 cw = 0x037F
  sp[1] = 0;                       // r15
  sp[2] = 0;                       // r14
  sp[3] = 0;                       // r13
  sp[4] = 0;                       // r12
  sp[5] = 0;                       // rbx
  sp[6] = 0;                       // rbp
  sp[7] = (uint64_t)(uintptr_t)fn; // ret target
  return sp;
}

}  // namespace bam
