-- brpc_amd: Wireshark dissector for the baidu_std wire protocol
-- (parity: reference tools/wireshark_baidu_std.lua — clean-room).
--
-- Wire layout (rpc/wire.h, policy/std_protocol.cc):
--   12-byte header: "PRPC" | body_size u32be | meta_size u32be
--   body = RpcMeta (protobuf) + payload (+ attachment)
-- Load via Wireshark: Analyze -> Lua -> Evaluate, or drop into the
-- plugins directory. Dissects the header and the RpcMeta envelope
-- (correlation id, service/method, compress type, attachment size).

local bam_proto = Proto("baidu_std_amd", "brpc_amd baidu_std RPC")

local f_magic = ProtoField.string("baidu_std_amd.magic", "Magic")
local f_body_size = ProtoField.uint32("baidu_std_amd.body_size", "Body size", base.DEC)
local f_meta_size = ProtoField.uint32("baidu_std_amd.meta_size", "Meta size", base.DEC)
local f_meta = ProtoField.bytes("baidu_std_amd.meta", "RpcMeta (protobuf)")
local f_payload = ProtoField.bytes("baidu_std_amd.payload", "Payload")
local f_cid = ProtoField.uint64("baidu_std_amd.correlation_id", "Correlation id")
local f_method = ProtoField.string("baidu_std_amd.method", "Service.Method")

bam_proto.fields = {f_magic, f_body_size, f_meta_size, f_meta, f_payload, f_cid, f_method}

-- Minimal protobuf walk of RpcMeta: only top-level varint/length fields
-- we need (field numbers per rpc/wire.h RpcMeta layout).
local function read_varint(tvb, off, maxoff)
  local v, shift = 0, 0
  while off < maxoff do
    local b = tvb(off, 1):uint()
    v = v + bit.lshift(bit.band(b, 0x7f), shift)
    off = off + 1
    if bit.band(b, 0x80) == 0 then return v, off end
    shift = shift + 7
  end
  return nil, off
end

function bam_proto.dissector(tvb, pinfo, tree)
  if tvb:len() < 12 then return 0 end
  if tvb(0, 4):string() ~= "PRPC" then return 0 end
  pinfo.cols.protocol = "BAIDU_STD"
  local body_size = tvb(4, 4):uint()
  local meta_size = tvb(8, 4):uint()
  local sub = tree:add(bam_proto, tvb(), "baidu_std (brpc_amd)")
  sub:add(f_magic, tvb(0, 4))
  sub:add(f_body_size, tvb(4, 4))
  sub:add(f_meta_size, tvb(8, 4))
  local have = tvb:len() - 12
  if have <= 0 then return 12 end
  local mlen = math.min(meta_size, have)
  local meta_tree = sub:add(f_meta, tvb(12, mlen))
  -- walk meta fields
  local off, maxoff = 12, 12 + mlen
  while off < maxoff do
    local tag
    tag, off = read_varint(tvb, off, maxoff)
    if tag == nil then break end
    local field = bit.rshift(tag, 3)
    local wt = bit.band(tag, 7)
    if wt == 0 then
      local v
      v, off = read_varint(tvb, off, maxoff)
      if v == nil then break end
      if field == 3 then  -- correlation_id
        meta_tree:add(f_cid, v)
      end
    elseif wt == 2 then
      local len
      len, off = read_varint(tvb, off, maxoff)
      if len == nil or off + len > maxoff then break end
      if field == 1 or field == 2 then  -- request/response meta submessage
        -- find service/method strings inside (fields 1/2, wt 2)
        local ioff, iend = off, off + len
        while ioff < iend do
          local itag
          itag, ioff = read_varint(tvb, ioff, iend)
          if itag == nil then break end
          local ifield = bit.rshift(itag, 3)
          local iwt = bit.band(itag, 7)
          if iwt == 2 then
            local ilen
            ilen, ioff = read_varint(tvb, ioff, iend)
            if ilen == nil or ioff + ilen > iend then break end
            if ifield == 1 or ifield == 2 then
              meta_tree:add(f_method, tvb(ioff, ilen))
            end
            ioff = ioff + ilen
          elseif iwt == 0 then
            local _
            _, ioff = read_varint(tvb, ioff, iend)
          else
            break
          end
        end
      end
      off = off + len
    else
      break
    end
  end
  local pay = have - mlen
  if pay > 0 then sub:add(f_payload, tvb(12 + mlen, pay)) end
  return 12 + have
end

-- Heuristic registration on TCP: matches the PRPC magic.
local function heuristic(tvb, pinfo, tree)
  if tvb:len() >= 4 and tvb(0, 4):string() == "PRPC" then
    bam_proto.dissector(tvb, pinfo, tree)
    return true
  end
  return false
end

bam_proto:register_heuristic("tcp", heuristic)
