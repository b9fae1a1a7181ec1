"""Data plane CPU tests: SYSHARD format, object store, mover,
input/output data through the executor."""
import os
import random

import pytest

from shipyard_amd.data import integrity, lz4py, mover, shardfmt
from shipyard_amd.data.storage import ObjectStore
from shipyard_amd.executor import LocalExecutor


class TestShardFmt:
    @pytest.mark.parametrize("n", [0, 1, 100, 64 * 1024, 200_000])
    def test_roundtrip(self, n):
        random.seed(n)
        data = bytes(random.choices(b"abcdefgh01234567", k=n))
        packed = shardfmt.pack(data)
        assert shardfmt.unpack_cpu(packed) == data

    def test_incompressible_stored(self):
        data = os.urandom(100_000)
        packed = shardfmt.pack(data)
        idx = shardfmt.read_index(packed)
        assert all(b.stored for b in idx.blocks)
        assert shardfmt.unpack_cpu(packed) == data

    def test_corruption_detected(self):
        data = b"hello world " * 10000
        packed = bytearray(shardfmt.pack(data))
        idx = shardfmt.read_index(bytes(packed))
        packed[idx.payload_off + 4] ^= 0xFF  # flip a block-0 payload byte
        with pytest.raises(ValueError):
            shardfmt.unpack_cpu(bytes(packed))

    def test_alignment(self):
        data = os.urandom(70_000)
        idx = shardfmt.read_index(shardfmt.pack(data))
        for b in idx.blocks:
            assert b.comp_off % 16 == 0


class TestIntegrity:
    def test_manifest_roundtrip(self):
        data = os.urandom(3 * (1 << 20) + 12345)
        m = integrity.compute_cpu(data)
        m2 = integrity.Manifest.from_json(m.to_json())
        assert integrity.verify(m, m2)

    def test_detects_change(self):
        data = bytearray(os.urandom(100_000))
        m = integrity.compute_cpu(bytes(data), chunk_size=65536)
        data[50] ^= 1
        m2 = integrity.compute_cpu(bytes(data), chunk_size=65536)
        assert not integrity.verify(m, m2)


class TestObjectStore:
    def test_roundtrip_plain(self, tmp_path):
        st = ObjectStore(tmp_path / "store")
        st.upload_bytes("c/dir/a.bin", b"hello", manifest=True)
        assert st.download_bytes("c/dir/a.bin") == b"hello"
        assert list(st.list("c")) == ["c/dir/a.bin"]

    def test_roundtrip_packed(self, tmp_path):
        st = ObjectStore(tmp_path / "store")
        data = b"compress me " * 5000
        st.upload_bytes("c/b.bin", data, pack=True)
        assert st.download_bytes("c/b.bin") == data

    def test_escape_rejected(self, tmp_path):
        st = ObjectStore(tmp_path / "store")
        with pytest.raises(ValueError):
            st.upload_bytes("../evil", b"x")

    def test_include_exclude(self, tmp_path):
        st = ObjectStore(tmp_path / "store")
        st.upload_bytes("c/a.png", b"x")
        st.upload_bytes("c/b.txt", b"x")
        assert list(st.list("c", include=["*.png"])) == ["c/a.png"]
        assert list(st.list("c", exclude=["*.png"])) == ["c/b.txt"]


class TestMover:
    def _mktree(self, root, sizes):
        for name, size in sizes.items():
            p = root / name
            p.parent.mkdir(parents=True, exist_ok=True)
            p.write_bytes(os.urandom(size))

    def test_ingress_parallel_split(self, tmp_path):
        src = tmp_path / "src"
        self._mktree(src, {"big.bin": 3 * (1 << 20), "sub/small.txt": 100})
        res = mover.ingress_directory(src, tmp_path / "dst", workers=3,
                                      split_mb=1, verify=True)
        assert res.files == 2
        assert res.verified
        assert (tmp_path / "dst" / "big.bin").read_bytes() == \
            (src / "big.bin").read_bytes()
        assert (tmp_path / "dst" / "sub" / "small.txt").exists()

    def test_bin_packing_balances(self, tmp_path):
        src = tmp_path / "s"
        self._mktree(src, {f"f{i}": 1000 * (i + 1) for i in range(10)})
        files = mover._gather_files(src, (), ())
        plan = mover._bin_pack(files, 3, None)
        loads = [sum(c[3] for c in b) for b in plan]
        assert max(loads) - min(loads) <= 10000

    def test_object_store_roundtrip(self, tmp_path):
        src = tmp_path / "src"
        self._mktree(src, {"a.bin": 500_000, "b/c.bin": 1000})
        st = ObjectStore(tmp_path / "store")
        res = mover.ingress_to_object_store(src, st, "ingested", pack=True)
        assert res.files == 2
        out = tmp_path / "out"
        res2 = mover.egress_from_object_store(st, "ingested", out)
        assert res2.files == 2
        assert (out / "a.bin").read_bytes() == (src / "a.bin").read_bytes()
        assert (out / "b" / "c.bin").read_bytes() == \
            (src / "b" / "c.bin").read_bytes()


class TestExecutorDataFlow:
    def test_input_output_data(self, tmp_path):
        ex = LocalExecutor(tmp_path / "root", detect_gpus=False)
        ex.pool_add({"pool_specification": {
            "id": "p", "gpus": {"dedicated": 0}, "cpu_slots": 2,
            "node_configuration": {"rocm": {"verify": False}}}})
        # seed the object store
        ex.stores["default"].upload_bytes("indata/seed.txt", b"seed-123\n")
        ex.jobs_add({"job_specifications": [{
            "id": "jdata",
            "tasks": [{
                "id": "t",
                "command": "cat seed.txt > result.txt; echo extra >> result.txt",
                "input_data": {"local_storage": [
                    {"remote_path": "indata"}]},
                "output_data": {"local_storage": [
                    {"remote_path": "outdata", "include": ["result.txt"]}]},
            }],
        }]}, "p")
        ex.run_until_idle(timeout=60)
        t = ex.tasks_list("jdata")[0]
        assert t["state"] == "completed"
        got = ex.stores["default"].download_bytes("outdata/result.txt")
        assert b"seed-123" in got and b"extra" in got
        ex.store.close()

    def test_task_output_chain(self, tmp_path):
        """local_batch input: task consumes a prior task's outputs
        (the cargo/task_file_mover analogue)."""
        ex = LocalExecutor(tmp_path / "root", detect_gpus=False)
        ex.pool_add({"pool_specification": {
            "id": "p", "gpus": {"dedicated": 0}, "cpu_slots": 2,
            "node_configuration": {"rocm": {"verify": False}}}})
        ex.jobs_add({"job_specifications": [{
            "id": "jchain",
            "tasks": [
                {"id": "producer", "command": "echo payload > out.dat"},
                {"id": "consumer",
                 "command": "grep payload out.dat",
                 "depends_on": ["producer"],
                 "input_data": {"local_batch": [
                     {"job_id": "jchain", "task_id": "producer",
                      "include": ["out.dat"]}]}},
            ],
        }]}, "p")
        ex.run_until_idle(timeout=60)
        states = {t["id"]: t["state"] for t in ex.tasks_list("jchain")}
        assert states == {"producer": "completed", "consumer": "completed"}
        ex.store.close()


class TestObjectStoreOps:
    def test_delete_and_exists(self, tmp_path):
        st = ObjectStore(tmp_path / "s")
        st.upload_bytes("c/x.bin", b"1")
        st.upload_bytes("c/y.bin", b"2")
        assert st.exists("c/x.bin")
        assert st.delete("c/x.bin")
        assert not st.exists("c/x.bin")
        assert not st.delete("c/x.bin")  # already gone
        assert st.delete("c")            # directory delete
        assert list(st.list("")) == []

    def test_manifest_files_hidden_from_list(self, tmp_path):
        st = ObjectStore(tmp_path / "s")
        st.upload_bytes("c/a.bin", b"x", manifest=True)
        assert list(st.list("c")) == ["c/a.bin"]

    def test_download_missing_raises(self, tmp_path):
        st = ObjectStore(tmp_path / "s")
        import pytest as _pytest

        with _pytest.raises(FileNotFoundError):
            st.download_bytes("nope.bin")


class TestParallelPack:
    def test_parallel_pack_identical_to_serial(self):
        import os
        import random

        random.seed(9)
        data = (b"layer content " * 40000 + os.urandom(200_000)
                + bytes(random.choices(b"qrstuv", k=150_000)))
        serial = shardfmt.pack(data)
        parallel = shardfmt.pack(data, workers=2)
        assert serial == parallel
        assert shardfmt.unpack_cpu(parallel) == data

    def test_parallel_pack_empty_and_tiny(self):
        assert shardfmt.pack(b"", workers=2) == shardfmt.pack(b"")
        assert shardfmt.pack(b"x", workers=2) == shardfmt.pack(b"x")

    def test_native_compressor_roundtrip_fuzz(self):
        """Native C++ LZ4 compressor output decodes with the python
        reference decoder for adversarial inputs."""
        import os
        import random

        from shipyard_amd import ops

        if not ops.native_compress_available():
            pytest.skip("ops library not built")
        random.seed(77)
        for trial in range(40):
            kind = trial % 4
            n = random.randrange(0, 20000)
            if kind == 0:
                data = os.urandom(n)
            elif kind == 1:
                data = bytes(random.choices(b"ab", k=n))
            elif kind == 2:
                data = (b"x" * random.randrange(1, 300)) * (n // 100 + 1)
            else:
                data = bytes(random.choices(range(256), k=n))
            comps = ops.lz4_compress_blocks(data, 4096)
            for i, c in enumerate(comps):
                raw = data[i * 4096:(i + 1) * 4096]
                if c is not None:
                    assert lz4py.decompress_block(c, len(raw)) == raw, \
                        (trial, i)
            # and the full shard path
            assert shardfmt.unpack_cpu(shardfmt.pack(data)) == data
