"""The hand-encoded TensorBoard event file must be a valid TFRecord
stream of Event protobufs — verified by an independent decoder here
(CRC32C framing + protobuf wire format), so a stock TensorBoard can
load what SummaryWriter writes."""

import glob
import struct

from sat_amd.utils.tb_events import TBEventWriter, _masked_crc


def _read_records(path):
    out = []
    with open(path, 'rb') as f:
        while True:
            hdr = f.read(8)
            if len(hdr) < 8:
                break
            (n,) = struct.unpack('<Q', hdr)
            (hcrc,) = struct.unpack('<I', f.read(4))
            assert hcrc == _masked_crc(hdr), 'length CRC mismatch'
            data = f.read(n)
            (dcrc,) = struct.unpack('<I', f.read(4))
            assert dcrc == _masked_crc(data), 'payload CRC mismatch'
            out.append(data)
    return out


def _parse_fields(buf):
    """Minimal protobuf wire-format parser -> {field_num: [values]}."""
    fields = {}
    i = 0
    while i < len(buf):
        key = 0
        shift = 0
        while True:
            b = buf[i]
            i += 1
            key |= (b & 0x7F) << shift
            shift += 7
            if not b & 0x80:
                break
        num, wire = key >> 3, key & 7
        if wire == 0:        # varint
            v = 0
            shift = 0
            while True:
                b = buf[i]
                i += 1
                v |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
        elif wire == 1:      # 64-bit
            v = struct.unpack('<d', buf[i:i + 8])[0]
            i += 8
        elif wire == 2:      # length-delimited
            ln = 0
            shift = 0
            while True:
                b = buf[i]
                i += 1
                ln |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
            v = buf[i:i + ln]
            i += ln
        elif wire == 5:      # 32-bit
            v = struct.unpack('<f', buf[i:i + 4])[0]
            i += 4
        else:
            raise AssertionError('wire type %d' % wire)
        fields.setdefault(num, []).append(v)
    return fields


def test_tb_event_file_round_trips(tmp_path):
    w = TBEventWriter(str(tmp_path))
    w.add_scalar('loss/total', 1.5, 7)
    w.add_scalars({'a': 2.0, 'b': -3.5}, 8)
    w.close()

    files = glob.glob(str(tmp_path / 'events.out.tfevents.*'))
    assert len(files) == 1
    recs = _read_records(files[0])
    assert len(recs) == 3

    # record 0: file_version
    f0 = _parse_fields(recs[0])
    assert f0[3][0] == b'brain.Event:2'

    # record 1: one scalar
    f1 = _parse_fields(recs[1])
    assert f1[2][0] == 7                      # step
    summ = _parse_fields(f1[5][0])
    val = _parse_fields(summ[1][0])
    assert val[1][0] == b'loss/total'
    assert abs(val[2][0] - 1.5) < 1e-6

    # record 2: two scalars in one summary
    f2 = _parse_fields(recs[2])
    assert f2[2][0] == 8
    summ = _parse_fields(f2[5][0])
    tags = {_parse_fields(v)[1][0]: _parse_fields(v)[2][0]
            for v in summ[1]}
    assert abs(tags[b'a'] - 2.0) < 1e-6
    assert abs(tags[b'b'] + 3.5) < 1e-6


def test_summary_writer_emits_both_formats(tmp_path):
    from sat_amd.utils.summary import SummaryWriter
    w = SummaryWriter(str(tmp_path))
    w.add_scalars({'total_loss': 1.0}, 1)
    w.close()
    assert (tmp_path / 'events.jsonl').exists()
    assert glob.glob(str(tmp_path / 'events.out.tfevents.*'))


def test_tb_histogram_round_trips(tmp_path):
    w = TBEventWriter(str(tmp_path))
    w.add_histogram('wts/hist', [-1.0, -0.5, 0.0, 0.25, 1.0, 1.0], 3)
    w.close()
    files = glob.glob(str(tmp_path / 'events.out.tfevents.*'))
    recs = _read_records(files[0])
    f = _parse_fields(recs[1])
    assert f[2][0] == 3
    summ = _parse_fields(f[5][0])
    val = _parse_fields(summ[1][0])
    assert val[1][0] == b'wts/hist'
    histo = _parse_fields(val[5][0])
    assert histo[1][0] == -1.0          # min
    assert histo[2][0] == 1.0           # max
    assert histo[3][0] == 6.0           # num
    assert abs(histo[4][0] - 0.75) < 1e-9   # sum
    # packed repeated doubles: counts sum to num
    counts = histo[7][0]
    import struct as st
    cs = [st.unpack('<d', counts[i:i + 8])[0]
          for i in range(0, len(counts), 8)]
    assert sum(cs) == 6.0
