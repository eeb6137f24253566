"""LLDB helper: backtrace suspended brpc_amd fibers (parity: reference
tools/lldb_bthread_stack.py; gdb flavor in tools/gdb_fiber_stack.py).

Usage inside lldb:
  (lldb) command script import tools/lldb_fiber_stack.py
  (lldb) fiber-bt <FiberMeta-address>

A suspended fiber's registers live at meta->ctx_sp (layout written by
src/fiber/context.S: fpu control word, r15, r14, r13, r12, rbx, rbp,
return address). The command temporarily rewrites rip/rsp/rbp of the
selected thread, prints the backtrace, then restores them.
"""
import shlex

import lldb


def fiber_bt(debugger, command, result, internal_dict):
    args = shlex.split(command)
    if len(args) != 1:
        print("usage: fiber-bt <FiberMeta-address>", file=result)
        return
    target = debugger.GetSelectedTarget()
    process = target.GetProcess()
    thread = process.GetSelectedThread()
    frame = thread.GetFrameAtIndex(0)

    meta = target.EvaluateExpression("(bam::FiberMeta*)%s" % args[0])
    sp = meta.GetChildMemberWithName("ctx_sp").GetValueAsUnsigned()
    err = lldb.SBError()
    rip = process.ReadPointerFromMemory(sp + 56, err)
    rbp = process.ReadPointerFromMemory(sp + 48, err)
    if err.Fail():
        print("cannot read context at %#x: %s" % (sp, err), file=result)
        return
    print("fiber sp=%#x rip=%#x rbp=%#x" % (sp, rip, rbp), file=result)

    regs = frame.GetRegisters().GetFirstValueByName("General Purpose Registers")
    save = {r: regs.GetChildMemberWithName(r).GetValueAsUnsigned()
            for r in ("rip", "rsp", "rbp")}
    for reg, val in (("rip", rip), ("rsp", sp + 64), ("rbp", rbp)):
        frame.FindRegister(reg).SetValueFromCString(hex(val), err)
    debugger.HandleCommand("thread backtrace")
    for reg, val in save.items():
        frame.FindRegister(reg).SetValueFromCString(hex(val), err)


def __lldb_init_module(debugger, internal_dict):
    debugger.HandleCommand(
        "command script add -f lldb_fiber_stack.fiber_bt fiber-bt")
    print("fiber-bt installed (see tools/lldb_fiber_stack.py)")
