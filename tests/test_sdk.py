"""Python SDK: fsspec filesystem + torch tensor reads."""
import os

import pytest

from curvine_amd.testing import SyncMiniCluster


@pytest.fixture
def cluster(tmp_path):
    smc = SyncMiniCluster(tmp_dir=str(tmp_path)).start()
    yield smc
    smc.stop()


def test_fsspec_roundtrip(cluster):
    import fsspec

    from curvine_amd.sdk.fsspec_fs import register
    register()
    master = f"127.0.0.1:{cluster.master.rpc.port}"
    fs = fsspec.filesystem("cv", master=master, skip_instance_cache=True)
    fs.mkdir("/sdk/dir")
    data = os.urandom(5 << 20)
    with fs.open("/sdk/dir/blob.bin", "wb") as f:
        f.write(data)
    assert fs.exists("/sdk/dir/blob.bin")
    info = fs.info("/sdk/dir/blob.bin")
    assert info["size"] == len(data) and info["type"] == "file"
    with fs.open("/sdk/dir/blob.bin", "rb") as f:
        assert f.read(100) == data[:100]
        f.seek(1 << 20)
        assert f.read(50) == data[1 << 20:(1 << 20) + 50]
    assert fs.cat_file("/sdk/dir/blob.bin", 10, 20) == data[10:20]
    names = fs.ls("/sdk/dir", detail=False)
    assert any(n.endswith("blob.bin") for n in names)
    fs.mv("/sdk/dir/blob.bin", "/sdk/moved.bin")
    assert fs.exists("/sdk/moved.bin")
    fs.rm("/sdk", recursive=True)
    assert not fs.exists("/sdk/moved.bin")


def test_pandas_over_fsspec(cluster):
    pd = pytest.importorskip("pandas")
    import fsspec

    from curvine_amd.sdk.fsspec_fs import register
    register()
    master = f"127.0.0.1:{cluster.master.rpc.port}"
    fs = fsspec.filesystem("cv", master=master, skip_instance_cache=True)
    with fs.open("/df.csv", "wb") as f:
        f.write(b"a,b\n1,2\n3,4\n")
    df = pd.read_csv(f"cv://df.csv", storage_options={
        "master": master, "skip_instance_cache": True})
    assert list(df.columns) == ["a", "b"] and len(df) == 2


def test_tensor_read_cpu(cluster):
    import torch

    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.torch_io import CurvineTensorReader
    sf = SyncFs(cluster.client_conf())
    data = os.urandom(2 << 20)
    sf.write_file("/tensor.bin", data)
    r = CurvineTensorReader(sf, "/tensor.bin")
    t = torch.zeros(len(data), dtype=torch.uint8)
    n = r.read_into_tensor(t)
    assert n == len(data)
    assert bytes(t.numpy().tobytes()) == data
    # offset read
    t2 = r.to_tensor(device="cpu", file_off=1000, n=500)
    assert t2.numpy().tobytes() == data[1000:1500]
    r.close()
    sf.shutdown()


@pytest.mark.gpu
def test_tensor_read_hbm_to_device(tmp_path):
    """HBM-cached file -> cuda tensor: pure device-to-device copy."""
    import torch

    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.torch_io import CurvineTensorReader
    from curvine_amd.testing import SyncMiniCluster, test_conf
    conf = test_conf(str(tmp_path))
    conf.worker.data_dirs = ["[HBM:512MB:0]gpu0"]
    smc = SyncMiniCluster(conf=conf, tmp_dir=str(tmp_path)).start()
    try:
        sf = SyncFs(smc.client_conf())
        data = os.urandom(32 << 20)
        sf.write_file("/hbm.bin", data, storage_tier="HBM")
        r = CurvineTensorReader(sf, "/hbm.bin")
        t = torch.zeros(len(data), dtype=torch.uint8, device="cuda:0")
        n = r.read_into_tensor(t)
        assert n == len(data)
        assert t.cpu().numpy().tobytes() == data
        r.close()
        sf.shutdown()
    finally:
        smc.stop()


def test_torch_dataloader_over_shards(cluster, tmp_path):
    """WebDataset-style tar shards through a torch DataLoader."""
    import io
    import tarfile

    import torch
    from torch.utils.data import DataLoader

    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.dataset import CurvineShardDataset
    sf = SyncFs(cluster.client_conf())
    samples = {}
    shard_paths = []
    for s in range(3):
        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w") as tf:
            for i in range(10):
                name = f"sample-{s}-{i}.bin"
                payload = os.urandom(1000 + i)
                samples[name] = payload
                info = tarfile.TarInfo(name)
                info.size = len(payload)
                tf.addfile(info, io.BytesIO(payload))
        path = f"/shards/shard-{s:03d}.tar"
        sf.write_file(path, buf.getvalue())
        shard_paths.append(path)
    ds = CurvineShardDataset(cluster.client_conf(), shard_paths)
    loader = DataLoader(ds, batch_size=None, num_workers=0)
    seen = {}
    for name, payload in loader:
        seen[name] = bytes(payload)
    assert seen == samples
    sf.shutdown()


def test_file_dataset(cluster):
    from curvine_amd.sdk.dataset import CurvineFileDataset
    from curvine_amd.client.filesystem import SyncFs
    sf = SyncFs(cluster.client_conf())
    for i in range(5):
        sf.write_file(f"/ds/item{i}.bin", bytes([i]) * 100)
    ds = CurvineFileDataset(cluster.client_conf(), "/ds")
    assert len(ds) == 5
    assert ds[2] == bytes([2]) * 100
    sf.shutdown()


def test_device_loader_cpu(cluster):
    """CurvineDeviceLoader on the MEM tier with a cpu tensor: tar headers
    indexed via short-circuit preads, payloads gathered host-side."""
    import io
    import tarfile

    import torch

    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.dataset import CurvineDeviceLoader

    sf = SyncFs(cluster.client_conf())
    samples = {}
    shard_paths = []
    for s in range(3):
        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w") as tf:
            for i in range(12):
                name = f"dl-{s}-{i}" + ("x" * 120 if i == 5 else "")
                payload = os.urandom(700 + 37 * i)
                samples[name] = payload
                info = tarfile.TarInfo(name)
                info.size = len(payload)
                tf.addfile(info, io.BytesIO(payload))
        path = f"/dl/shard-{s:03d}.tar"
        sf.write_file(path, buf.getvalue())
        shard_paths.append(path)

    dl = CurvineDeviceLoader(cluster.client_conf(), shard_paths,
                             device="cpu", batch_size=7, shuffle=True, seed=3)
    assert dl.num_samples == len(samples)
    seen = {}
    for tensor, sections, names in dl:
        assert tensor.dtype == torch.uint8
        for (start, ln), name in zip(sections, names):
            seen[name] = bytes(tensor[start:start + ln].numpy().tobytes())
    assert seen == samples
    dl.close()
    sf.shutdown()


def test_object_store_and_safe_commit(cluster):
    """curvine-lancedb analog: object-store surface + conditional-put
    commit (atomic create is the linearization point)."""
    from curvine_amd import errors as err
    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.object_store import (CommitConflict,
                                              ConditionalPutCommitter,
                                              CurvineObjectStore)

    sf = SyncFs(cluster.client_conf())
    store = CurvineObjectStore(sf, prefix="/tables")
    store.put("t1/data/part-0.bin", b"D" * 5000)
    store.put("t1/data/part-1.bin", b"E" * 100)
    assert store.head("t1/data/part-0.bin").size == 5000
    assert store.get("t1/data/part-0.bin", 10, 20) == b"D" * 20
    keys = [m.key for m in store.list("t1")]
    assert keys == ["t1/data/part-0.bin", "t1/data/part-1.bin"]
    dirs, objs = store.list_with_delimiter("t1")
    assert dirs == ["t1/data"] and objs == []

    # create mode: second put of the same key must lose
    store.put("t1/lock", b"w1", mode="create")
    with pytest.raises(err.FileAlreadyExists):
        store.put("t1/lock", b"w2", mode="create")
    assert store.get("t1/lock") == b"w1"

    # copy / rename
    store.copy("t1/data/part-1.bin", "t1/data/part-1.copy")
    assert store.get("t1/data/part-1.copy") == b"E" * 100
    store.rename("t1/data/part-1.copy", "t1/data/part-2.bin")
    assert not store.exists("t1/data/part-1.copy")

    # multipart writer
    with store.put_multipart("t1/big.bin") as w:
        for _ in range(8):
            w.write(b"x" * (1 << 20))
    assert store.head("t1/big.bin").size == 8 << 20

    # conditional-put commit handler
    c = ConditionalPutCommitter(store, "t1")
    assert c.latest_version() == 0
    c.commit(1, b"manifest-v1")
    c.commit(2, b"manifest-v2")
    with pytest.raises(CommitConflict):
        c.commit(2, b"manifest-v2-loser")
    assert c.latest_version() == 2
    assert c.read_manifest(2) == b"manifest-v2"
    sf.shutdown()


def test_pyarrow_parquet_dataset(cluster):
    """Arrow analytics on cv://: write a partitioned parquet dataset
    through fsspec, read it back with pyarrow.dataset."""
    pa = pytest.importorskip("pyarrow")
    import pyarrow.dataset as pads
    import pyarrow.parquet as pq
    import fsspec

    from curvine_amd.sdk.fsspec_fs import register
    register()
    master = f"127.0.0.1:{cluster.master.rpc.port}"
    fs = fsspec.filesystem("cv", master=master, skip_instance_cache=True)
    table = pa.table({"k": list(range(1000)),
                      "part": [i % 4 for i in range(1000)],
                      "v": [float(i) * 0.5 for i in range(1000)]})
    pq.write_to_dataset(table, "/pq/ds", partition_cols=["part"],
                        filesystem=fs)
    ds = pads.dataset("/pq/ds", filesystem=fs, partitioning="hive")
    got = ds.to_table()
    assert got.num_rows == 1000
    assert sorted(got.column("k").to_pylist()) == list(range(1000))
    filt = ds.to_table(filter=pads.field("part") == 2)
    assert filt.num_rows == 250
