import os
import struct

from transformer_amd.runtime.summary import SummaryWriter, _masked_crc
from transformer_amd.config import parse_flags, flags_dict


def _read_records(path):
    records = []
    with open(path, "rb") as f:
        while True:
            hdr = f.read(8)
            if len(hdr) < 8:
                break
            (length,) = struct.unpack("<Q", hdr)
            (crc_h,) = struct.unpack("<I", f.read(4))
            assert crc_h == _masked_crc(hdr)
            data = f.read(length)
            (crc_d,) = struct.unpack("<I", f.read(4))
            assert crc_d == _masked_crc(data)
            records.append(data)
    return records


def test_event_file_framing(tmp_path):
    w = SummaryWriter(str(tmp_path))
    w.add_scalar("loss", 1.5, step=3)
    w.add_scalar("accuracy", 0.25, step=3)
    w.close()
    files = [f for f in os.listdir(tmp_path) if f.startswith("events.out.tfevents")]
    assert len(files) == 1
    recs = _read_records(os.path.join(tmp_path, files[0]))
    assert len(recs) == 3  # file_version + 2 scalars
    assert b"brain.Event:2" in recs[0]
    assert b"loss" in recs[1]
    assert struct.pack("<f", 1.5) in recs[1]
    assert b"accuracy" in recs[2]


def test_flag_defaults_match_reference():
    # reference utils.py:18-33 defaults
    args = parse_flags([])
    d = flags_dict(args)
    assert d["dataset_path"] == "data/"
    assert d["buffer_size"] == 100000
    assert d["src_vocab_file"] == "src_vocab.txt"
    assert d["tgt_vocab_file"] == "tgt_vocab.txt"
    assert d["sequence_length"] == 50
    assert d["epochs"] == 4
    assert d["batch_size"] == 64
    assert d["per_replica_batch_size"] == 16
    assert d["num_layers"] == 4
    assert d["d_model"] == 512
    assert d["dff"] == 1024
    assert d["num_heads"] == 4
    assert d["enable_function"] is True
    assert d["max_ckpt_keep"] == 5
    assert d["ckpt_path"] == "model_dist"
    assert d["dropout_rate"] == 0.1
    assert d["warmup_steps"] == 60000  # Q3
    assert d["label_smoothing"] == 0.0  # Q9 default = reference numerics


def test_flag_forms():
    args = parse_flags(["--d_model=256", "--noenable_function",
                        "--batch_size", "32"])
    assert args.d_model == 256
    assert args.enable_function is False
    assert args.batch_size == 32
