"""GPU tests: shard stager pipeline (pinned double-buffer -> HBM ->
decode -> verify) vs CPU references."""
import os
import random

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

from shipyard_amd.data import shardfmt  # noqa: E402
from shipyard_amd.data.stager import ShardStager  # noqa: E402


@pytest.mark.parametrize("native", [True, False],
                         ids=["native-cxx", "python"])
def test_stage_shard_file(tmp_path, native):
    random.seed(5)
    data = (bytes(random.choices(b"abcdefgh", k=500_000)) +
            os.urandom(300_000))
    src = tmp_path / "shard.syshard"
    src.write_bytes(shardfmt.pack(data))
    st = ShardStager(staging_mb=1, verify=True, native=native)
    tensor, res = st.stage_file(src)
    torch.cuda.synchronize()
    assert res.decoded and res.verified
    assert res.raw_bytes == len(data)
    assert bytes(tensor.cpu().numpy().tobytes()) == data


def test_stage_plain_file(tmp_path):
    data = os.urandom(3_000_000)
    src = tmp_path / "plain.bin"
    src.write_bytes(data)
    st = ShardStager(staging_mb=1)
    tensor, res = st.stage_file(src)
    torch.cuda.synchronize()
    assert not res.decoded
    assert bytes(tensor.cpu().numpy().tobytes()) == data


def test_stage_detects_corruption(tmp_path):
    data = b"all work and no play " * 50_000
    packed = bytearray(shardfmt.pack(data))
    idx = shardfmt.read_index(bytes(packed))
    # corrupt a stored/compressed block payload byte
    packed[idx.payload_off + 2] ^= 0xAA
    src = tmp_path / "bad.syshard"
    src.write_bytes(bytes(packed))
    st = ShardStager(staging_mb=1, verify=True)
    with pytest.raises(ValueError):
        st.stage_file(src)
