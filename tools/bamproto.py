#!/usr/bin/env python3
"""bamproto — C++ stub generator for brpc_amd (≙ reference protoc plugin
integration: protoc-generated google::protobuf stubs on Channel/Server,
brpc/channel.h:189-228 / server.cpp:844-875, and mcpack2pb/generator.cpp's
role as a codegen tool).

Usage: python tools/bamproto.py input.proto > out.bam.h

Parses the .proto with the framework's own DescriptorPool (base/proto.*)
and emits a self-contained header: one struct per message with typed
fields + ParseFromString/SerializeToString (standard protobuf wire), one
<Service>Base class whose RegisterTo(Server*) adapts typed virtual
methods onto the byte-level Service registry, and one <Service>_Stub
whose typed methods call any ChannelBase (plain Channel, ParallelChannel,
SelectiveChannel...)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import brpc_amd as b  # noqa: E402

P = b.core.proto

# FieldDef::Type enum order (base/proto.h)
(T_DOUBLE, T_FLOAT, T_INT32, T_INT64, T_UINT32, T_UINT64, T_SINT32, T_SINT64,
 T_FIXED32, T_FIXED64, T_SFIXED32, T_SFIXED64, T_BOOL, T_STRING, T_BYTES,
 T_ENUM, T_MESSAGE) = range(17)

CPP_TYPE = {
    T_DOUBLE: "double", T_FLOAT: "float", T_INT32: "int32_t", T_INT64: "int64_t",
    T_UINT32: "uint32_t", T_UINT64: "uint64_t", T_SINT32: "int32_t",
    T_SINT64: "int64_t", T_FIXED32: "uint32_t", T_FIXED64: "uint64_t",
    T_SFIXED32: "int32_t", T_SFIXED64: "int64_t", T_BOOL: "bool",
    T_STRING: "std::string", T_BYTES: "std::string", T_ENUM: "int32_t",
}

VARINTS = {T_INT32, T_INT64, T_UINT32, T_UINT64, T_BOOL, T_ENUM}
ZIGZAGS = {T_SINT32, T_SINT64}
FIX64 = {T_FIXED64, T_SFIXED64}
FIX32 = {T_FIXED32, T_SFIXED32}


def cpp_name(full, pkg):
    """test.pb.Inner -> Inner (nested: Outer.Inner -> Outer_Inner)."""
    n = full[len(pkg) + 1:] if pkg and full.startswith(pkg + ".") else full
    return n.replace(".", "_")


def wire_type(t):
    if t in VARINTS or t in ZIGZAGS:
        return 0
    if t == T_DOUBLE or t in FIX64:
        return 1
    if t == T_FLOAT or t in FIX32:
        return 5
    return 2


def emit_value_write(out, f, expr, indent):
    t = f["type"]
    num = f["number"]
    i = " " * indent
    if t in VARINTS:
        out.append(f"{i}bam::pbgen::put_tag(out, {num}, 0);")
        out.append(f"{i}bam::pbgen::put_varint(out, (uint64_t){expr});")
    elif t in ZIGZAGS:
        out.append(f"{i}bam::pbgen::put_tag(out, {num}, 0);")
        out.append(f"{i}bam::pbgen::put_varint(out, bam::pbgen::zigzag_enc((int64_t){expr}));")
    elif t == T_DOUBLE:
        out.append(f"{i}bam::pbgen::put_tag(out, {num}, 1);")
        out.append(f"{i}bam::pbgen::put_double(out, {expr});")
    elif t == T_FLOAT:
        out.append(f"{i}bam::pbgen::put_tag(out, {num}, 5);")
        out.append(f"{i}bam::pbgen::put_float(out, {expr});")
    elif t in FIX64:
        out.append(f"{i}bam::pbgen::put_tag(out, {num}, 1);")
        out.append(f"{i}bam::pbgen::put_fixed64(out, (uint64_t){expr});")
    elif t in FIX32:
        out.append(f"{i}bam::pbgen::put_tag(out, {num}, 5);")
        out.append(f"{i}bam::pbgen::put_fixed32(out, (uint32_t){expr});")
    elif t in (T_STRING, T_BYTES):
        out.append(f"{i}bam::pbgen::put_len_delim(out, {num}, {expr});")
    else:  # message
        out.append(f"{i}{{ std::string sub; {expr}.SerializeToString(&sub); "
                   f"bam::pbgen::put_len_delim(out, {num}, sub); }}")


def emit_value_read(out, f, target, indent, pkg):
    t = f["type"]
    i = " " * indent
    if t in VARINTS:
        out.append(f"{i}{{ uint64_t _vv; if (!bam::pbgen::get_varint(p, end, &_vv)) return false; "
                   f"{target} = ({CPP_TYPE[t] if t != T_BOOL else 'bool'})"
                   f"{'(_vv != 0)' if t == T_BOOL else '(_vv)'}; }}")
    elif t in ZIGZAGS:
        out.append(f"{i}{{ uint64_t _vv; if (!bam::pbgen::get_varint(p, end, &_vv)) return false; "
                   f"{target} = ({CPP_TYPE[t]})bam::pbgen::zigzag_dec(_vv); }}")
    elif t == T_DOUBLE:
        out.append(f"{i}{{ uint64_t _vv; if (!bam::pbgen::get_fixed64(p, end, &_vv)) return false; "
                   f"memcpy(&{target}, &_vv, 8); }}")
    elif t == T_FLOAT:
        out.append(f"{i}{{ uint32_t _vv; if (!bam::pbgen::get_fixed32(p, end, &_vv)) return false; "
                   f"float _fv; memcpy(&_fv, &_vv, 4); {target} = _fv; }}")
    elif t in FIX64:
        out.append(f"{i}{{ uint64_t _vv; if (!bam::pbgen::get_fixed64(p, end, &_vv)) return false; "
                   f"{target} = ({CPP_TYPE[t]})_vv; }}")
    elif t in FIX32:
        out.append(f"{i}{{ uint32_t _vv; if (!bam::pbgen::get_fixed32(p, end, &_vv)) return false; "
                   f"{target} = ({CPP_TYPE[t]})_vv; }}")
    elif t in (T_STRING, T_BYTES):
        out.append(f"{i}if (!bam::pbgen::get_len_delim(p, end, &{target})) return false;")
    else:  # message
        out.append(f"{i}{{ std::string sub; if (!bam::pbgen::get_len_delim(p, end, &sub)) "
                   f"return false; if (!{target}.ParseFromBytes(sub.data(), sub.size())) "
                   f"return false; }}")


class Out(list):
    """A list that is also callable (append) — emit helpers take either."""
    def __call__(self, line):
        self.append(line)


def generate(proto_path):
    src = open(proto_path).read()
    pool = P.Pool()
    pool.parse(src)
    # package = longest common dotted prefix of message names' first parts:
    # recover from the source text instead.
    pkg = ""
    for line in src.splitlines():
        line = line.strip()
        if line.startswith("package "):
            pkg = line[len("package "):].rstrip(";").strip()
            break

    msgs = pool.messages()
    enums = pool.enums()
    out = Out()
    o = out
    o("// Generated by tools/bamproto.py from %s — DO NOT EDIT." % os.path.basename(proto_path))
    o("#pragma once")
    o("")
    o("#include <map>")
    o("#include <memory>")
    o("#include <string>")
    o("#include <vector>")
    o("")
    o('#include "base/iobuf.h"')
    o('#include "base/mcpack.h"')
    o('#include "base/pbgen.h"')
    o('#include "rpc/channel.h"')
    o('#include "rpc/closure.h"')
    o('#include "rpc/controller.h"')
    o('#include "rpc/server.h"')
    o("")
    ns = pkg.replace(".", "::") if pkg else "bamgen"
    o("namespace %s {" % ns)
    o("")
    for en in enums:
        name = cpp_name(en, pkg)
        o("enum %s : int32_t {" % name)
        for vname, vnum in sorted(pool.enum_values(en).items(), key=lambda kv: kv[1]):
            o("  %s = %d," % (vname, vnum))
        o("};")
        o("")
    # forward decls
    for msg in msgs:
        if msg.endswith("Entry"):
            continue
        o("struct %s;" % cpp_name(msg, pkg))
    o("")
    for msg in msgs:
        if msg.endswith("Entry"):
            continue  # map entries are emitted as std::map fields
        emit_message(o, pool, msg, pkg)
    for svc in pool.services():
        emit_service(o, pool, svc, pkg)
    o("}  // namespace %s" % ns)
    return "\n".join(out) + "\n"


def field_cpp_type(f, pkg, pool):
    t = f["type"]
    if f["is_map"]:
        entry = pool.describe_message(f["type_name"])
        kf = [x for x in entry if x["number"] == 1][0]
        vf = [x for x in entry if x["number"] == 2][0]
        kt = CPP_TYPE.get(kf["type"], "std::string")
        vt = (cpp_name(vf["type_name"], pkg) if vf["type"] == T_MESSAGE
              else CPP_TYPE.get(vf["type"], "int64_t"))
        return "std::map<%s, %s>" % (kt, vt)
    if t == T_MESSAGE:
        base = cpp_name(f["type_name"], pkg)
        return "std::vector<%s>" % base if f["repeated"] else base
    base = CPP_TYPE[t]
    return "std::vector<%s>" % base if f["repeated"] else base


def emit_message(o, pool, msg, pkg):
    name = cpp_name(msg, pkg)
    fields = pool.describe_message(msg)
    o("struct %s {" % name)
    for f in fields:
        ft = field_cpp_type(f, pkg, pool)
        init = "" if (f["repeated"] or f["is_map"] or
                      f["type"] in (T_STRING, T_BYTES, T_MESSAGE)) else " = %s" % (
            "false" if f["type"] == T_BOOL else "0")
        o("  %s %s%s;" % (ft, f["name"], init))
    o("  std::string _unknown;  // unrecognized fields, re-emitted verbatim")
    o("")
    o("  bool ParseFromBytes(const char* p, size_t n) {")
    o("    const char* end = p + n;")
    o("    while (p < end) {")
    o("      const char* tag_start = p;")
    o("      uint64_t tag;")
    o("      if (!bam::pbgen::get_varint(p, end, &tag)) return false;")
    o("      const int wt = (int)(tag & 7);")
    o("      switch ((int)(tag >> 3)) {")
    for f in fields:
        o("        case %d: {" % f["number"])
        emit_field_parse(o, pool, f, pkg)
        o("          break;")
        o("        }")
    o("        default:")
    o("          if (!bam::pbgen::skip_field(p, end, wt)) return false;")
    o("          _unknown.append(tag_start, p - tag_start);")
    o("      }")
    o("      (void)tag_start;")
    o("    }")
    o("    return true;")
    o("  }")
    o("  bool ParseFromString(const std::string& s) { return ParseFromBytes(s.data(), s.size()); }")
    o("")
    o("  void SerializeToString(std::string* out) const {")
    for f in fields:
        emit_field_serialize(o, pool, f, pkg)
    o("    out->append(_unknown);")
    o("  }")
    o("  std::string SerializeAsString() const { std::string s; SerializeToString(&s); return s; }")
    o("")
    emit_mcpack(o, pool, fields, pkg)
    o("};")
    o("")


def mcpack_value_expr(f, expr):
    """C++ expression building a bam::mcpack::Value from a scalar field."""
    t = f["type"]
    if t == T_BOOL:
        return "bam::mcpack::Value::Bool(%s)" % expr
    if t in (T_UINT32, T_UINT64, T_FIXED32, T_FIXED64):
        return "bam::mcpack::Value::Uint((uint64_t)%s)" % expr
    if t in (T_DOUBLE, T_FLOAT):
        return "bam::mcpack::Value::Double((double)%s)" % expr
    if t == T_STRING:
        return "bam::mcpack::Value::Str(%s)" % expr
    if t == T_BYTES:
        return "bam::mcpack::Value::Bin(%s)" % expr
    return "bam::mcpack::Value::Int((int64_t)%s)" % expr


def mcpack_read_stmt(o, f, src, dst, indent):
    """Emit assignment of mcpack Value `src` into scalar field `dst`."""
    t = f["type"]
    i = " " * indent
    if t == T_BOOL:
        o(f"{i}{dst} = {src}.type == bam::mcpack::Value::BOOL ? {src}.b : ({src}.i != 0);")
    elif t in (T_UINT32, T_UINT64, T_FIXED32, T_FIXED64):
        o(f"{i}{dst} = ({CPP_TYPE[t]})({src}.type == bam::mcpack::Value::UINT ? {src}.u : (uint64_t){src}.i);")
    elif t in (T_DOUBLE, T_FLOAT):
        o(f"{i}{dst} = ({CPP_TYPE[t]})({src}.type == bam::mcpack::Value::DOUBLE ? {src}.d : (double){src}.i);")
    elif t in (T_STRING, T_BYTES):
        o(f"{i}{dst} = {src}.str;")
    else:
        o(f"{i}{dst} = ({CPP_TYPE[t]})({src}.type == bam::mcpack::Value::UINT ? (int64_t){src}.u : {src}.i);")


def emit_mcpack(o, pool, fields, pkg):
    """mcpack (de)serializers on generated structs — the reference's
    protoc-gen-mcpack role (mcpack2pb/generator.cpp)."""
    o("  // ---- mcpack v2 (parity: protoc-gen-mcpack generated converters) ----")
    o("  void ToMcpackValue(bam::mcpack::Value* out) const {")
    o("    *out = bam::mcpack::Value::Object();")
    for f in fields:
        name = f["name"]
        if f["is_map"]:
            entry = pool.describe_message(f["type_name"])
            vf = [x for x in entry if x["number"] == 2][0]
            o("    { bam::mcpack::Value m = bam::mcpack::Value::Object();")
            o("      for (const auto& kv : %s) {" % name)
            if vf["type"] == T_MESSAGE:
                o("        bam::mcpack::Value mv; kv.second.ToMcpackValue(&mv);")
            else:
                o("        bam::mcpack::Value mv = %s;" % mcpack_value_expr(vf, "kv.second"))
            kf = [x for x in entry if x["number"] == 1][0]
            if kf["type"] == T_STRING:
                o("        m.obj[kv.first] = mv;")
            else:
                o("        m.obj[std::to_string(kv.first)] = mv;")
            o("      }")
            o("      out->obj[\"%s\"] = m; }" % name)
        elif f["repeated"]:
            o("    { bam::mcpack::Value a = bam::mcpack::Value::Array();")
            o("      for (const auto& x : %s) {" % name)
            if f["type"] == T_MESSAGE:
                o("        bam::mcpack::Value e; x.ToMcpackValue(&e); a.arr.push_back(e);")
            else:
                o("        a.arr.push_back(%s);" % mcpack_value_expr(f, "x"))
            o("      }")
            o("      out->obj[\"%s\"] = a; }" % name)
        elif f["type"] == T_MESSAGE:
            o("    { bam::mcpack::Value m; %s.ToMcpackValue(&m); out->obj[\"%s\"] = m; }"
              % (name, name))
        else:
            o("    out->obj[\"%s\"] = %s;" % (name, mcpack_value_expr(f, name)))
    o("  }")
    o("  bool FromMcpackValue(const bam::mcpack::Value& v) {")
    o("    if (v.type != bam::mcpack::Value::OBJECT) return false;")
    for f in fields:
        name = f["name"]
        o("    { auto it = v.obj.find(\"%s\");" % name)
        o("      if (it != v.obj.end()) {")
        if f["is_map"]:
            entry = pool.describe_message(f["type_name"])
            kf = [x for x in entry if x["number"] == 1][0]
            vf = [x for x in entry if x["number"] == 2][0]
            o("        for (const auto& kv : it->second.obj) {")
            if kf["type"] == T_STRING:
                key_expr = "kv.first"
            else:
                key_expr = "(%s)strtoll(kv.first.c_str(), nullptr, 10)" % CPP_TYPE[kf["type"]]
            if vf["type"] == T_MESSAGE:
                vt = cpp_name(vf["type_name"], pkg)
                o("          %s mv; mv.FromMcpackValue(kv.second); %s[%s] = mv;"
                  % (vt, name, key_expr))
            else:
                o("          %s mv{};" % CPP_TYPE[vf["type"]])
                mcpack_read_stmt(o, vf, "kv.second", "mv", 10)
                o("          %s[%s] = mv;" % (name, key_expr))
            o("        }")
        elif f["repeated"]:
            o("        for (const auto& e : it->second.arr) {")
            if f["type"] == T_MESSAGE:
                o("          %s.emplace_back();" % name)
                o("          if (!%s.back().FromMcpackValue(e)) return false;" % name)
            else:
                o("          %s x{};" % CPP_TYPE[f["type"]])
                mcpack_read_stmt(o, f, "e", "x", 10)
                o("          %s.push_back(x);" % name)
            o("        }")
        elif f["type"] == T_MESSAGE:
            o("        if (!%s.FromMcpackValue(it->second)) return false;" % name)
        else:
            mcpack_read_stmt(o, f, "it->second", name, 8)
        o("      } }")
    o("    return true;")
    o("  }")
    o("  bool SerializeAsMcpack(std::string* out) const {")
    o("    bam::mcpack::Value v;")
    o("    ToMcpackValue(&v);")
    o("    return bam::mcpack::Serialize(v, out);")
    o("  }")
    o("  bool ParseFromMcpack(const std::string& data) {")
    o("    bam::mcpack::Value v;")
    o("    if (!bam::mcpack::Parse(data.data(), data.size(), &v)) return false;")
    o("    return FromMcpackValue(v);")
    o("  }")


def emit_field_parse(o, pool, f, pkg):
    t = f["type"]
    if f["is_map"]:
        entry = pool.describe_message(f["type_name"])
        kf = [x for x in entry if x["number"] == 1][0]
        vf = [x for x in entry if x["number"] == 2][0]
        o("          std::string sub;")
        o("          if (!bam::pbgen::get_len_delim(p, end, &sub)) return false;")
        o("          const char* ep = sub.data(); const char* eend = ep + sub.size();")
        kt = CPP_TYPE.get(kf["type"], "std::string")
        vt = (cpp_name(vf["type_name"], pkg) if vf["type"] == T_MESSAGE
              else CPP_TYPE.get(vf["type"], "int64_t"))
        o("          %s k{}; %s v{};" % (kt, vt))
        o("          while (ep < eend) {")
        o("            uint64_t etag;")
        o("            if (!bam::pbgen::get_varint(ep, eend, &etag)) return false;")
        o("            if ((etag >> 3) == 1) {")
        sub = []
        emit_value_read(sub, kf, "k", 14, pkg)
        for l in sub:
            o(l.replace("(p, end,", "(ep, eend,"))
        o("            } else if ((etag >> 3) == 2) {")
        sub = []
        emit_value_read(sub, vf, "v", 14, pkg)
        for l in sub:
            o(l.replace("(p, end,", "(ep, eend,"))
        o("            } else if (!bam::pbgen::skip_field(ep, eend, (int)(etag & 7))) {")
        o("              return false;")
        o("            }")
        o("          }")
        o("          %s[k] = v;" % f["name"])
        return
    if f["repeated"]:
        if t not in (T_STRING, T_BYTES, T_MESSAGE):
            # accept both packed and unpacked encodings
            o("          if (wt == 2) {")
            o("            std::string sub;")
            o("            if (!bam::pbgen::get_len_delim(p, end, &sub)) return false;")
            o("            const char* rp = sub.data(); const char* rend = rp + sub.size();")
            o("            while (rp < rend) {")
            o("              %s x{};" % CPP_TYPE[t])
            sub = []
            emit_value_read(sub, f, "x", 14, pkg)
            for l in sub:
                o(l.replace("(p, end,", "(rp, rend,"))
            o("              %s.push_back(x);" % f["name"])
            o("            }")
            o("          } else {")
            o("            %s x{};" % CPP_TYPE[t])
            sub = []
            emit_value_read(sub, f, "x", 12, pkg)
            for l in sub:
                o(l)
            o("            %s.push_back(x);" % f["name"])
            o("          }")
        elif t == T_MESSAGE:
            o("          %s.emplace_back();" % f["name"])
            emit_value_read(o, f, "%s.back()" % f["name"], 10, pkg)
        else:
            o("          %s.emplace_back();" % f["name"])
            emit_value_read(o, f, "%s.back()" % f["name"], 10, pkg)
        return
    emit_value_read(o, f, f["name"], 10, pkg)


def emit_field_serialize(o, pool, f, pkg):
    t = f["type"]
    name = f["name"]
    if f["is_map"]:
        entry = pool.describe_message(f["type_name"])
        kf = [x for x in entry if x["number"] == 1][0]
        vf = [x for x in entry if x["number"] == 2][0]
        o("    for (const auto& kv : %s) {" % name)
        o("      std::string e;")
        sub = []
        emit_value_write(sub, {**kf, "number": 1}, "kv.first", 6)
        emit_value_write(sub, {**vf, "number": 2}, "kv.second", 6)
        for l in sub:
            o(l.replace("(out,", "(&e,"))
        o("      bam::pbgen::put_len_delim(out, %d, e);" % f["number"])
        o("    }")
        return
    if f["repeated"]:
        if f["packed"] and t not in (T_STRING, T_BYTES, T_MESSAGE):
            o("    if (!%s.empty()) {" % name)
            o("      std::string packed;")
            sub = []
            emit_value_write(sub, f, "x", 6)
            # strip the per-element tag for packed runs
            body = [l.replace("(out,", "(&packed,") for l in sub if "put_tag" not in l]
            o("      for (const auto& x : %s) {" % name)
            for l in body:
                o("  " + l)
            o("      }")
            o("      bam::pbgen::put_len_delim(out, %d, packed);" % f["number"])
            o("    }")
        else:
            o("    for (const auto& x : %s) {" % name)
            emit_value_write(o, f, "x", 6)
            o("    }")
        return
    # singular: proto3 omits default values (messages: we track by empty
    # serialization? emit always for message fields with any content)
    if t in (T_STRING, T_BYTES):
        o("    if (!%s.empty()) {" % name)
        emit_value_write(o, f, name, 6)
        o("    }")
    elif t == T_MESSAGE:
        o("    { std::string sub; %s.SerializeToString(&sub);" % name)
        o("      if (!sub.empty()) bam::pbgen::put_len_delim(out, %d, sub); }" % f["number"])
    elif t == T_BOOL:
        o("    if (%s) {" % name)
        emit_value_write(o, f, name, 6)
        o("    }")
    elif t in (T_DOUBLE, T_FLOAT):
        o("    if (%s != 0) {" % name)
        emit_value_write(o, f, name, 6)
        o("    }")
    else:
        o("    if (%s != 0) {" % name)
        emit_value_write(o, f, name, 6)
        o("    }")


def emit_service(o, pool, svc, pkg):
    name = cpp_name(svc, pkg)
    methods = pool.service_methods(svc)
    o("// ---- service %s ----" % svc)
    o("")
    o("class %sBase {" % name)
    o(" public:")
    o("  virtual ~%sBase() {}" % name)
    for mname, intype, outtype in methods:
        o("  virtual void %s(bam::Controller* cntl, const %s* request, %s* response,"
          % (mname, cpp_name(intype, pkg), cpp_name(outtype, pkg)))
        o("                  bam::Closure* done) = 0;")
    o("")
    o("  // Adapts the typed methods onto the byte-level Service registry")
    o("  // (requests parse into typed structs; done serializes responses).")
    o("  int RegisterTo(bam::Server* server) {")
    o("    auto* svc = new bam::Service(\"%s\");" % name)
    for mname, intype, outtype in methods:
        itn = cpp_name(intype, pkg)
        otn = cpp_name(outtype, pkg)
        o("    svc->AddMethod(\"%s\", [this](bam::Controller* cntl, const bam::IOBuf& req," % mname)
        o("                             bam::IOBuf* resp, bam::Closure* done) {")
        o("      auto* ctx = new std::pair<%s, %s>();" % (itn, otn))
        o("      std::string bytes = req.to_string();")
        o("      if (!ctx->first.ParseFromString(bytes)) {")
        o("        delete ctx;")
        o("        cntl->SetFailed(1003 /*EREQUEST*/, \"malformed %s\");" % itn)
        o("        done->Run();")
        o("        return;")
        o("      }")
        o("      bam::Closure* wrapped = bam::NewCallback([ctx, resp, done] {")
        o("        std::string out;")
        o("        ctx->second.SerializeToString(&out);")
        o("        resp->append(out);")
        o("        delete ctx;")
        o("        done->Run();")
        o("      });")
        o("      this->%s(cntl, &ctx->first, &ctx->second, wrapped);" % mname)
        o("    });")
    o("    return server->AddService(svc, bam::SERVER_OWNS_SERVICE);")
    o("  }")
    o("};")
    o("")
    o("class %s_Stub {" % name)
    o(" public:")
    o("  explicit %s_Stub(bam::ChannelBase* channel) : channel_(channel) {}" % name)
    o("")
    for mname, intype, outtype in methods:
        itn = cpp_name(intype, pkg)
        otn = cpp_name(outtype, pkg)
        o("  void %s(bam::Controller* cntl, const %s* request, %s* response," % (mname, itn, otn))
        o("          bam::Closure* done) {")
        o("    bam::IOBuf req_buf;")
        o("    std::string bytes;")
        o("    request->SerializeToString(&bytes);")
        o("    req_buf.append(bytes);")
        o("    auto* resp_buf = new bam::IOBuf;")
        o("    bam::Closure* wrapped = bam::NewCallback([resp_buf, response, cntl, done] {")
        o("      if (!cntl->Failed()) {")
        o("        std::string out = resp_buf->to_string();")
        o("        if (!response->ParseFromString(out))")
        o("          cntl->SetFailed(1007 /*ERESPONSE*/, \"malformed %s\");" % otn)
        o("      }")
        o("      delete resp_buf;")
        o("      if (done != nullptr) done->Run();")
        o("    });")
        o("    if (done == nullptr) {")
        o("      // synchronous: CallMethod blocks; run the parse inline")
        o("      channel_->CallMethod(\"%s.%s\", cntl, &req_buf, resp_buf, nullptr);" % (name, mname))
        o("      wrapped->Run();")
        o("    } else {")
        o("      channel_->CallMethod(\"%s.%s\", cntl, &req_buf, resp_buf, wrapped);" % (name, mname))
        o("    }")
        o("  }")
    o("")
    o(" private:")
    o("  bam::ChannelBase* channel_;")
    o("};")
    o("")


if __name__ == "__main__":
    if len(sys.argv) != 2:
        print(__doc__)
        sys.exit(1)
    sys.stdout.write(generate(sys.argv[1]))
