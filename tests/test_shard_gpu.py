"""GPU test: SYSHARD unpack (LZ4 decode + CRC verify) on the MI355X
vs the CPU reference reader."""
import os
import random

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

from shipyard_amd.data import shardfmt  # noqa: E402
from shipyard_amd.data.integrity import compute_cpu, compute_gpu, verify  # noqa: E402


def test_unpack_gpu_matches_cpu():
    random.seed(11)
    data = (bytes(random.choices(b"abcdefgh", k=300_000)) +
            os.urandom(150_000) + b"Z" * 90_000)
    packed = shardfmt.pack(data)
    got = shardfmt.unpack_gpu(packed)
    torch.cuda.synchronize()
    assert bytes(got.cpu().numpy().tobytes()) == data


def test_unpack_gpu_detects_corruption():
    data = b"the quick brown fox " * 20_000
    packed = bytearray(shardfmt.pack(data))
    idx = shardfmt.read_index(bytes(packed))
    packed[idx.payload_off + 8] ^= 0x55
    with pytest.raises(ValueError):
        shardfmt.unpack_gpu(bytes(packed))


def test_gpu_manifest_matches_cpu():
    data = os.urandom(2 * (1 << 20) + 777)
    d = torch.frombuffer(bytearray(data), dtype=torch.uint8).to("cuda:0")
    m_gpu = compute_gpu(d, chunk_size=65536)
    m_cpu = compute_cpu(data, chunk_size=65536)
    assert verify(m_cpu, m_gpu)
    assert m_cpu.sha256_root == m_gpu.sha256_root
