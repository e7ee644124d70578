import glob
import os
import struct

import numpy as np

from cyclegan_amd.utils.tb_writer import EventWriter, crc32c, _masked_crc
from cyclegan_amd.utils import Summary


def test_crc32c_known_vectors():
    # RFC 3720 test vector: 32 bytes of zeros -> 0x8a9136aa
    assert crc32c(b"\x00" * 32) == 0x8A9136AA
    assert crc32c(b"123456789") == 0xE3069283


def _read_records(path):
    recs = []
    with open(path, "rb") as f:
        while True:
            hdr = f.read(8)
            if len(hdr) < 8:
                break
            (n,) = struct.unpack("<Q", hdr)
            (crc_h,) = struct.unpack("<I", f.read(4))
            assert crc_h == _masked_crc(hdr)
            data = f.read(n)
            (crc_d,) = struct.unpack("<I", f.read(4))
            assert crc_d == _masked_crc(data)
            recs.append(data)
    return recs


def test_event_file_roundtrip(tmp_path):
    w = EventWriter(str(tmp_path))
    w.scalar("loss_G/total", 1.5, step=3)
    w.close()
    files = glob.glob(os.path.join(str(tmp_path), "events.out.tfevents.*"))
    assert len(files) == 1
    recs = _read_records(files[0])
    assert len(recs) == 2  # file_version + scalar
    assert b"brain.Event:2" in recs[0]
    assert b"loss_G/total" in recs[1]
    # float 1.5 little-endian appears in the scalar record
    assert struct.pack("<f", 1.5) in recs[1]


def test_summary_two_writer_layout(tmp_path):
    s = Summary(str(tmp_path))
    s.scalar("tag_train", 1.0, step=0, training=True)
    s.scalar("tag_test", 2.0, step=0, training=False)
    s.scalar("elapse", 3.0, step=0)  # default -> test writer (reference quirk)
    train_files = glob.glob(os.path.join(str(tmp_path), "events.*"))
    test_files = glob.glob(os.path.join(str(tmp_path), "test", "events.*"))
    assert len(train_files) == 1 and len(test_files) == 1
    train_data = open(train_files[0], "rb").read()
    test_data = open(test_files[0], "rb").read()
    assert b"tag_train" in train_data and b"tag_test" not in train_data
    assert b"elapse" in test_data


def test_summary_image(tmp_path):
    s = Summary(str(tmp_path))
    img = (np.random.rand(2, 8, 8, 3) * 255).astype("uint8")
    s.image("X_cycle/sample", img, step=1, training=False)
    test_files = glob.glob(os.path.join(str(tmp_path), "test", "events.*"))
    data = open(test_files[0], "rb").read()
    assert b"\x89PNG" in data


def test_plot_cycle_emits_cycle_panels(tmp_path):
    """plot_cycle (reference utils.py:112-145): runs cycle_step over the
    plot pairs and writes X_cycle/Y_cycle image panels to the test
    writer with per-sample tags."""
    import argparse
    import torch
    from cyclegan_amd.parallel import DistContext
    from cyclegan_amd.trainer import CycleGAN
    from cyclegan_amd.utils import plot_cycle

    a = argparse.Namespace(output_dir=str(tmp_path), batch_size=1,
                           global_batch_size=1, num_residual_blocks=1,
                           compute_dtype=torch.float32)
    torch.manual_seed(0)
    gan = CycleGAN(a, DistContext(device=torch.device("cpu")))
    s = Summary(str(tmp_path))
    pairs = [(torch.rand(1, 64, 64, 3) * 2 - 1,
              torch.rand(1, 64, 64, 3) * 2 - 1) for _ in range(2)]
    plot_cycle(pairs, gan, s, epoch=0)
    s.close()
    test_files = glob.glob(os.path.join(str(tmp_path), "test",
                                        "events.out.tfevents.*"))
    assert test_files
    blob = b"".join(_read_records(test_files[0]))
    assert b"X_cycle" in blob and b"Y_cycle" in blob
