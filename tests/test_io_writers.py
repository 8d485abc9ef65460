"""Streaming fast file writers (ref deepspeed/io/)."""
import os

import torch

from deepspeed_amd.io import (FastFileWriter, MockFileWriter, PyFileWriter,
                              save_with_fast_writer)


def test_mock_and_py_writers(tmp_path):
    m = MockFileWriter(str(tmp_path / "m"))
    m.write(b"x" * 100)
    assert m.fini()["write_bytes"] == 100
    p = str(tmp_path / "p.bin")
    with PyFileWriter(p) as w:
        w.write(b"hello")
    assert open(p, "rb").read() == b"hello"


def test_fast_writer_roundtrip_small(tmp_path):
    """Tail-only path (payload smaller than one buffer half)."""
    p = str(tmp_path / "f.bin")
    with FastFileWriter(p, buffer_bytes=8192) as w:
        w.write(b"abc" * 100)
    assert open(p, "rb").read() == b"abc" * 100


def test_fast_writer_multibuffer(tmp_path):
    """Payload crossing several drain boundaries, odd tail."""
    p = str(tmp_path / "big.bin")
    data = bytes(range(256)) * 300 + b"tail!"
    w = FastFileWriter(p, buffer_bytes=8192)
    for i in range(0, len(data), 1000):  # misaligned chunks
        w.write(data[i:i + 1000])
    w.close()
    got = open(p, "rb").read()
    assert len(got) == len(data) and got == data
    st = w.stats
    assert st["write_bytes"] == len(data)
    assert st["fill_count"] == len(data) // 4096


def test_fast_writer_with_aio_handle(tmp_path):
    """Drains ride the aio offset API when the ext is present."""
    from deepspeed_amd.ops.loader import get_ext
    ext = get_ext(required=False)
    if ext is None:
        import pytest
        pytest.skip("ext not built")
    h = ext.aio_handle(1 << 16, 4, False, False, 2)
    p = str(tmp_path / "aio.bin")
    data = os.urandom(40000)
    with FastFileWriter(p, handle=h, buffer_bytes=8192) as w:
        w.write(data)
    assert open(p, "rb").read() == data
    assert w.stats["aio_bytes"] > 0


def test_torch_save_through_fast_writer(tmp_path):
    p = str(tmp_path / "ckpt.pt")
    obj = {"w": torch.randn(1000, 64), "step": 7,
           "nested": {"b": torch.arange(100)}}
    stats = save_with_fast_writer(obj, p, buffer_bytes=32768)
    back = torch.load(p, weights_only=False)
    assert back["step"] == 7
    assert torch.equal(back["w"], obj["w"])
    assert torch.equal(back["nested"]["b"], obj["nested"]["b"])
    assert stats["write_bytes"] == os.path.getsize(p)


def test_fast_writer_single_buffer_aio(tmp_path):
    """double_buffer=False: drain must complete before the (single)
    buffer is refilled — data integrity across many refills."""
    from deepspeed_amd.ops.loader import get_ext
    ext = get_ext(required=False)
    if ext is None:
        import pytest
        pytest.skip("ext not built")
    h = ext.aio_handle(1 << 16, 4, False, False, 2)
    p = str(tmp_path / "s.bin")
    data = os.urandom(50000)
    w = FastFileWriter(p, handle=h, buffer_bytes=8192, double_buffer=False)
    for i in range(0, len(data), 1234):
        w.write(data[i:i + 1234])
    w.close()
    assert open(p, "rb").read() == data


def test_fast_writer_randomized_chunks(tmp_path):
    """Property-style: random chunk sequences across buffer boundaries
    always reproduce the byte stream exactly."""
    import random
    rng = random.Random(7)
    for trial in range(6):
        p = str(tmp_path / f"r{trial}.bin")
        total = rng.randrange(1, 60000)
        data = bytes(rng.getrandbits(8) for _ in range(total))
        w = FastFileWriter(p, buffer_bytes=8192,
                           double_buffer=bool(trial % 2))
        i = 0
        while i < total:
            n = rng.randrange(1, 5000)
            w.write(data[i:i + n])
            i += n
        w.close()
        assert open(p, "rb").read() == data, (trial, total)
