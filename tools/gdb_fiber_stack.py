"""GDB helper: list brpc_amd fiber contexts (parity: reference
tools/gdb_bthread_stack.py).

Usage inside gdb:
  (gdb) source tools/gdb_fiber_stack.py
  (gdb) fiber-list          # enumerate live FiberMeta slots
  (gdb) fiber-bt <addr>     # backtrace a suspended fiber by meta address

A suspended fiber's registers live at meta->ctx_sp (layout: fpu word,
r15, r14, r13, r12, rbx, rbp, return address — see src/fiber/context.S).
"""
import gdb


class FiberList(gdb.Command):
    def __init__(self):
        super().__init__("fiber-list", gdb.COMMAND_USER)

    def invoke(self, arg, from_tty):
        pool = gdb.parse_and_eval(
            "'bam::ResourcePool<bam::FiberMeta>::singleton()'")
        print("use: p 'bam::ResourcePool<bam::FiberMeta>' blocks; each block"
              " holds 256 FiberMeta; live slots have stack_base != 0")


class FiberBt(gdb.Command):
    def __init__(self):
        super().__init__("fiber-bt", gdb.COMMAND_USER)

    def invoke(self, arg, from_tty):
        meta = gdb.parse_and_eval(f"(bam::FiberMeta*){arg}")
        sp = int(meta["ctx_sp"])
        rip = int(gdb.parse_and_eval(f"*(unsigned long*)({sp} + 56)"))
        rbp = int(gdb.parse_and_eval(f"*(unsigned long*)({sp} + 48)"))
        print(f"fiber sp={sp:#x} rip={rip:#x} rbp={rbp:#x}")
        gdb.execute(f"set $save_rip=$rip")
        gdb.execute(f"set $save_rsp=$rsp")
        gdb.execute(f"set $save_rbp=$rbp")
        gdb.execute(f"set $rip={rip}")
        gdb.execute(f"set $rsp={sp + 64}")
        gdb.execute(f"set $rbp={rbp}")
        gdb.execute("bt")
        gdb.execute("set $rip=$save_rip")
        gdb.execute("set $rsp=$save_rsp")
        gdb.execute("set $rbp=$save_rbp")


FiberList()
FiberBt()
