"""MPEG-TS muxer + HLS playlist tests (≙ reference brpc/ts.cpp TsWriter,
test coverage modeled on its FLV->TS remux path).

The test builds synthetic FLV AVC/AAC tags (AVCC sequence header + frames,
AudioSpecificConfig + raw AAC), muxes to TS, then PARSES the TS back with
an independent python TS reader: sync bytes, PAT->PMT chain, PES
reassembly, Annex B NALU payload round-trip and ADTS framing.
"""
import struct

import brpc_amd as b

r = b.core.rpc


def avc_seq_header(sps, pps):
    # AVCDecoderConfigurationRecord with 4-byte NALU lengths
    rec = bytes([1, sps[1], sps[2], sps[3], 0xFF, 0xE1])
    rec += struct.pack(">H", len(sps)) + sps
    rec += bytes([1]) + struct.pack(">H", len(pps)) + pps
    return bytes([0x17, 0x00, 0, 0, 0]) + rec


def avc_frame(nalus, keyframe, ct=0):
    body = bytes([0x17 if keyframe else 0x27, 0x01]) + ct.to_bytes(3, "big")
    for n in nalus:
        body += struct.pack(">I", len(n)) + n
    return body


def aac_seq_header():
    # AAC-LC (objectType 2), 44.1 kHz (index 4), stereo (2)
    asc = (2 << 11) | (4 << 7) | (2 << 3)
    return bytes([0xAF, 0x00]) + struct.pack(">H", asc)


def aac_frame(payload):
    return bytes([0xAF, 0x01]) + payload


def parse_ts(doc):
    """Minimal independent TS reader: returns ({pid: es_bytes}, pids_seen,
    pmt_streams)."""
    assert len(doc) % 188 == 0
    pes_acc = {}
    pids = set()
    pmt_streams = {}
    pmt_pid = None
    for off in range(0, len(doc), 188):
        pkt = doc[off:off + 188]
        assert pkt[0] == 0x47, "lost sync at offset %d" % off
        pusi = bool(pkt[1] & 0x40)
        pid = ((pkt[1] & 0x1F) << 8) | pkt[2]
        pids.add(pid)
        afc = (pkt[3] >> 4) & 0x3
        p = 4
        if afc & 0x2:  # adaptation field
            p += 1 + pkt[4]
        if not (afc & 0x1):
            continue  # no payload
        payload = pkt[p:]
        if pid == 0 and pusi:  # PAT
            sec = payload[1 + payload[0]:]
            assert sec[0] == 0x00
            sec_len = ((sec[1] & 0x0F) << 8) | sec[2]
            # single program: last 4 pre-CRC bytes are prog loop entry
            prog = sec[8:3 + sec_len - 4]
            pmt_pid = ((prog[2] & 0x1F) << 8) | prog[3]
        elif pid == pmt_pid and pusi:  # PMT
            sec = payload[1 + payload[0]:]
            assert sec[0] == 0x02
            sec_len = ((sec[1] & 0x0F) << 8) | sec[2]
            pil = ((sec[10] & 0x0F) << 8) | sec[11]
            i = 12 + pil
            end = 3 + sec_len - 4
            while i < end:
                stype = sec[i]
                spid = ((sec[i + 1] & 0x1F) << 8) | sec[i + 2]
                esl = ((sec[i + 3] & 0x0F) << 8) | sec[i + 4]
                pmt_streams[spid] = stype
                i += 5 + esl
        else:
            pes_acc.setdefault(pid, []).append((pusi, payload))
    # reassemble PES payloads (concatenate per pid, strip PES headers)
    es = {}
    for pid, chunks in pes_acc.items():
        units = []
        cur = b""
        for pusi, payload in chunks:
            if pusi:
                if cur:
                    units.append(cur)
                cur = payload
            else:
                cur += payload
        if cur:
            units.append(cur)
        out = b""
        for u in units:
            assert u[:3] == b"\x00\x00\x01"
            hdr_len = u[8]
            out += u[9 + hdr_len:]
        es[pid] = out
    return es, pids, pmt_streams


SPS = bytes([0x67, 0x42, 0x00, 0x1E, 0xAB, 0xCD])
PPS = bytes([0x68, 0xCE, 0x38, 0x80])


def make_flv():
    tags = [
        (9, 0, avc_seq_header(SPS, PPS)),
        (8, 0, aac_seq_header()),
        (9, 0, avc_frame([bytes([0x65]) + bytes(range(200))], True, ct=40)),
        (8, 12, aac_frame(bytes([0x21] * 64))),
        (9, 40, avc_frame([bytes([0x41]) + bytes(range(100))], False)),
        (8, 35, aac_frame(bytes([0x22] * 300))),
        (9, 80, avc_frame([bytes([0x41]) + bytes(197)], False)),
    ]
    flv_doc = b.core.rpc.flv_build([(t, ts, p) for t, ts, p in tags])
    return flv_doc, tags


def test_ts_structure_and_roundtrip():
    flv_doc, tags = make_flv()
    ts_doc = r.flv_to_ts(flv_doc)
    es, pids, streams = parse_ts(ts_doc)
    # PAT(0) + PMT + both elementary pids present
    assert 0 in pids and 0x1000 in pids
    assert streams == {0x100: 0x1B, 0x101: 0x0F}
    # video ES: Annex B with SPS/PPS re-injected before the keyframe
    v = es[0x100]
    assert v.startswith(b"\x00\x00\x00\x01" + SPS)
    assert b"\x00\x00\x00\x01" + PPS in v
    assert bytes([0x65]) + bytes(range(200)) in v
    assert bytes([0x41]) + bytes(range(100)) in v
    # audio ES: ADTS sync per frame, frame lengths cover header+payload
    a = es[0x101]
    assert a[0] == 0xFF and (a[1] & 0xF0) == 0xF0
    flen = ((a[3] & 0x03) << 11) | (a[4] << 3) | (a[5] >> 5)
    assert flen == 7 + 64
    assert a[7:7 + 64] == bytes([0x21] * 64)
    # second ADTS frame directly follows
    a2 = a[flen:]
    assert a2[0] == 0xFF and (a2[1] & 0xF0) == 0xF0


def test_ts_packet_alignment_and_counters():
    flv_doc, _ = make_flv()
    ts_doc = r.flv_to_ts(flv_doc)
    assert len(ts_doc) % 188 == 0
    # continuity counters increase mod 16 per pid over payload packets
    last = {}
    for off in range(0, len(ts_doc), 188):
        pkt = ts_doc[off:off + 188]
        pid = ((pkt[1] & 0x1F) << 8) | pkt[2]
        afc = (pkt[3] >> 4) & 0x3
        cc = pkt[3] & 0x0F
        if not (afc & 1):
            continue
        if pid in last:
            assert cc == (last[pid] + 1) % 16, "cc break on pid %#x" % pid
        last[pid] = cc


def test_ts_mux_tags_binding():
    _, tags = make_flv()
    ts_doc = r.ts_mux_tags([(t, ts, p) for t, ts, p in tags])
    es, _, streams = parse_ts(ts_doc)
    assert 0x100 in es and 0x101 in es and len(streams) == 2


def test_hls_playlist():
    m3u8 = r.hls_playlist([("seg0.ts", 9.984), ("seg1.ts", 10.0)], 10, 0, True)
    assert m3u8.startswith("#EXTM3U")
    assert "#EXT-X-TARGETDURATION:10" in m3u8
    assert "#EXTINF:9.984,\nseg0.ts" in m3u8
    assert m3u8.rstrip().endswith("#EXT-X-ENDLIST")
