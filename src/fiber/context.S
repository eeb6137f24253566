/* brpc_amd fiber context switch, x86_64 System V.
 * Clean-room implementation of an fcontext-style switch (capability parity
 * with reference bthread/context.cpp): saves callee-saved GPRs + mxcsr/x87
 * control words on the current stack, swaps rsp, restores, returns.
 *
 *   void* bam_jump_context(void** from_sp, void* to_sp, void* arg);
 *     rdi = where to store current sp, rsi = sp to switch to, rdx = arg.
 *     Returns (in rax) the arg passed by whoever later jumps back; on the
 *     first entry into a fresh context, control "returns" into the entry
 *     function with rdi = arg.
 */
  .text
  .globl bam_jump_context
  .type  bam_jump_context,@function
  .align 16
bam_jump_context:
  pushq %rbp
  pushq %rbx
  pushq %r12
  pushq %r13
  pushq %r14
  pushq %r15
  subq  $8, %rsp
  stmxcsr (%rsp)
  fnstcw  4(%rsp)
  movq  %rsp, (%rdi)      /* save current sp */
  movq  %rsi, %rsp        /* switch stacks */
  ldmxcsr (%rsp)
  fldcw   4(%rsp)
  addq  $8, %rsp
  popq  %r15
  popq  %r14
  popq  %r13
  popq  %r12
  popq  %rbx
  popq  %rbp
  movq  %rdx, %rax        /* value delivered to the resumed context */
  movq  %rdx, %rdi        /* first argument for a fresh context's entry fn */
  ret
  .size bam_jump_context,.-bam_jump_context
  .section .note.GNU-stack,"",@progbits
