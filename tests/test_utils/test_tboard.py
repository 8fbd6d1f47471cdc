"""TensorBoard event-file writer: the produced tfevents records must parse
(framing CRCs + protobuf wire format) without the tensorboard package."""

import struct

from sheeprl_amd.utils.tboard import TensorBoardWriter, _masked_crc


def _read_records(path):
    data = open(path, "rb").read()
    recs = []
    off = 0
    while off < len(data):
        (ln,) = struct.unpack_from("<Q", data, off)
        (lcrc,) = struct.unpack_from("<I", data, off + 8)
        assert lcrc == _masked_crc(data[off : off + 8])
        payload = data[off + 12 : off + 12 + ln]
        (pcrc,) = struct.unpack_from("<I", data, off + 12 + ln)
        assert pcrc == _masked_crc(payload)
        recs.append(payload)
        off += 12 + ln + 4
    return recs


def test_tfevents_roundtrip(tmp_path):
    w = TensorBoardWriter(str(tmp_path))
    w.add_scalar("Loss/total", 1.5, step=3)
    w.add_scalars({"Rewards/rew_avg": 10.0, "Time/sps": 2.25}, step=7)
    w.close()
    files = list(tmp_path.glob("events.out.tfevents.*"))
    assert len(files) == 1
    recs = _read_records(files[0])
    assert len(recs) == 4  # file_version + 3 scalars
    # first record carries the brain.Event:2 version string
    assert b"brain.Event:2" in recs[0]
    assert b"Loss/total" in recs[1]
    # simple_value 1.5 encoded little-endian float after tag field
    assert struct.pack("<f", 1.5) in recs[1]
    assert b"Rewards/rew_avg" in recs[2] and b"Time/sps" in recs[3]


def test_known_crc32c_vectors():
    # RFC 3720 test vector: 32 zero bytes -> CRC32C 0x8A9136AA
    from sheeprl_amd.utils.tboard import _crc32c

    assert _crc32c(b"\x00" * 32) == 0x8A9136AA
    assert _crc32c(b"123456789") == 0xE3069283
