"""Minimal TensorBoard event writer: framing + proto round-trip checks."""

import struct

from distributedmnist_amd.utils.tbwriter import (EventWriter, _crc32c,
                                                 _masked_crc, _varint)


def test_crc32c_known_vectors():
    # RFC 3720 test vectors
    assert _crc32c(b"") == 0x00000000
    assert _crc32c(b"123456789") == 0xE3069283
    assert _crc32c(bytes(32)) == 0x8A9136AA


def test_event_file_framing(tmp_path):
    w = EventWriter(str(tmp_path))
    w.add_scalar("Validation Accuracy", 0.91, 30)
    w.add_scalar("Validation Loss", 0.31, 30)
    w.close()
    files = list(tmp_path.glob("events.out.tfevents.*"))
    assert len(files) == 1
    data = files[0].read_bytes()
    # walk the TFRecord framing, verifying both CRCs of every record
    off, nrec = 0, 0
    while off < len(data):
        (ln,) = struct.unpack_from("<Q", data, off)
        (lcrc,) = struct.unpack_from("<I", data, off + 8)
        assert lcrc == _masked_crc(data[off:off + 8])
        rec = data[off + 12:off + 12 + ln]
        (dcrc,) = struct.unpack_from("<I", data, off + 12 + ln)
        assert dcrc == _masked_crc(rec)
        off += 12 + ln + 4
        nrec += 1
    assert nrec == 3  # file_version + 2 scalars
    assert b"brain.Event:2" in data
    assert b"Validation Accuracy" in data


def test_trainer_summarize_writes_events(tmp_path):
    from distributedmnist_amd.engine.train import Trainer, make_dataset
    from distributedmnist_amd.utils.flags import build_train_parser
    flags = build_train_parser().parse_args(
        ["--synthetic_data", "--train_dir", str(tmp_path),
         "--batch_size", "8", "--max_steps", "3", "--model", "mlp",
         "--device", "cpu", "--should_summarize",
         "--save_summaries_secs", "0", "--save_interval_secs", "100000"])
    t = Trainer(flags)
    ds = make_dataset(flags, 0, 1, t.device, t.compute_dtype)
    t.train(ds)
    assert list(tmp_path.glob("events.out.tfevents.*"))
