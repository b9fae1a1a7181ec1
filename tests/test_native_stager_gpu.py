"""GPU test: native C++ staging pipeline (sy_stage_file)."""
import os

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

from shipyard_amd import ops  # noqa: E402


@pytest.mark.parametrize("direct", [True, False], ids=["odirect", "buffered"])
def test_stage_file_native_roundtrip(tmp_path, direct):
    data = os.urandom(7 * (1 << 20) + 12345)
    src = tmp_path / "blob.bin"
    src.write_bytes(data)
    dst = torch.zeros(len(data), dtype=torch.uint8, device="cuda:0")
    secs = ops.stage_file_native(src, dst, staging_mb=2, use_direct=direct)
    torch.cuda.synchronize()
    assert secs > 0
    assert bytes(dst.cpu().numpy().tobytes()) == data


def test_stage_file_native_offset(tmp_path):
    data = os.urandom(1 << 20)
    src = tmp_path / "blob.bin"
    src.write_bytes(data)
    n = 300_000
    off = 4096
    dst = torch.zeros(n, dtype=torch.uint8, device="cuda:0")
    ops.stage_file_native(src, dst, file_off=off, n_bytes=n, staging_mb=1)
    torch.cuda.synchronize()
    assert bytes(dst.cpu().numpy().tobytes()) == data[off:off + n]


def test_stage_file_native_missing_file(tmp_path):
    dst = torch.zeros(16, dtype=torch.uint8, device="cuda:0")
    with pytest.raises(RuntimeError):
        ops.stage_file_native(tmp_path / "nope.bin", dst, n_bytes=16)
