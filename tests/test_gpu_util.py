"""gpu_info (mocked amd-smi/rocm-smi) + util + hdfs_path tests."""

import json
import os

import pytest

from tensorflowonspark_amd import TFNode, gpu_info, util


AMD_SMI_LIST = json.dumps([{"gpu": i, "bdf": "0000:0{}:00.0".format(i)}
                           for i in range(8)])
AMD_SMI_PROC_BUSY = json.dumps([
    {"gpu": 0, "process_list": [{"process_info": {"name": "python", "pid": 123}}]},
    {"gpu": 1, "process_list": []},
])
ROCM_SMI_ID = json.dumps({"card{}".format(i): {"GPU ID": "0x74b9"} for i in range(4)})


def _fake_run(table):
    def run(cmd):
        return table.get(cmd[0])
    return run


def test_list_and_busy(monkeypatch):
    monkeypatch.setattr(gpu_info, "_run",
                        _fake_run({"amd-smi": AMD_SMI_LIST}))
    assert gpu_info._list_gpu_ids() == list(range(8))
    assert gpu_info.is_gpu_available()


def test_rocm_smi_fallback(monkeypatch):
    table = {"amd-smi": None, "rocm-smi": ROCM_SMI_ID}
    monkeypatch.setattr(gpu_info, "_run", _fake_run(table))
    assert gpu_info._list_gpu_ids() == [0, 1, 2, 3]


def test_no_gpu(monkeypatch):
    monkeypatch.setattr(gpu_info, "_run", _fake_run({}))
    assert not gpu_info.is_gpu_available()


def test_get_gpus_deterministic_slice(monkeypatch):
    monkeypatch.setattr(gpu_info, "_run", _fake_run({"amd-smi": AMD_SMI_LIST}))
    monkeypatch.setattr(gpu_info, "_busy_gpu_ids", lambda: set())
    # worker i takes slice [i*num : (i+1)*num] of the free list
    assert gpu_info.get_gpus(2, 0) == ["0", "1"]
    assert gpu_info.get_gpus(2, 1) == ["2", "3"]
    assert gpu_info.get_gpus(2, 3) == ["6", "7"]
    # modulo wraparound
    assert gpu_info.get_gpus(2, 4) == ["0", "1"]
    assert gpu_info.get_gpus(1, 2, format=str) == "2"


def test_get_gpus_excludes_busy(monkeypatch):
    monkeypatch.setattr(gpu_info, "_run", _fake_run({"amd-smi": AMD_SMI_LIST}))
    monkeypatch.setattr(gpu_info, "_busy_gpu_ids", lambda: {0, 2})
    got = gpu_info.get_gpus(2, 0)
    assert got == ["1", "3"]


def test_get_gpus_insufficient_raises(monkeypatch):
    monkeypatch.setattr(gpu_info, "_run", _fake_run({"amd-smi": AMD_SMI_LIST}))
    monkeypatch.setattr(gpu_info, "_busy_gpu_ids", lambda: set(range(8)))
    monkeypatch.setattr(gpu_info.time, "sleep", lambda s: None)
    with pytest.raises(RuntimeError):
        gpu_info.get_gpus(1, 0)


def test_executor_id_roundtrip(tmp_path):
    util.write_executor_id(7, str(tmp_path))
    assert util.read_executor_id(str(tmp_path)) == 7


def test_executor_id_missing(tmp_path):
    with pytest.raises(RuntimeError):
        util.read_executor_id(str(tmp_path))


def test_find_in_path(tmp_path):
    f = tmp_path / "prog"
    f.write_text("x")
    path = os.pathsep.join(["/nonexistent", str(tmp_path)])
    assert util.find_in_path(path, "prog") == str(f)
    assert util.find_in_path(path, "missing") is False


class _Ctx:
    def __init__(self, fs, wd="/tmp/wd"):
        self.defaultFS = fs
        self.working_dir = wd


def test_hdfs_path_matrix():
    # known schemes pass through
    for scheme in ("hdfs://nn/", "file:///x/", "s3://b/", "viewfs://x/"):
        assert TFNode.hdfs_path(_Ctx("hdfs://nn"), scheme + "p") == scheme + "p"
    # absolute path -> defaultFS prefix
    assert TFNode.hdfs_path(_Ctx("hdfs://nn:8020"), "/data/x") == "hdfs://nn:8020/data/x"
    # relative on hdfs -> user home
    p = TFNode.hdfs_path(_Ctx("hdfs://nn:8020"), "rel")
    assert p.startswith("hdfs://nn:8020/user/") and p.endswith("/rel")
    # relative on local fs -> cwd
    assert TFNode.hdfs_path(_Ctx("file://", "/tmp/wd"), "rel") == "file:///tmp/wd/rel"


def test_host_peer_index():
    from tensorflowonspark_amd.TFSparkNode import _host_peer_index
    info = [{"host": "h1", "executor_id": 3}, {"host": "h1", "executor_id": 0},
            {"host": "h2", "executor_id": 1}, {"host": "h1", "executor_id": 5}]
    # sorted peers on h1: [0, 3, 5] -> deterministic disjoint slice indices
    assert _host_peer_index(info, "h1", 0) == 0
    assert _host_peer_index(info, "h1", 3) == 1
    assert _host_peer_index(info, "h1", 5) == 2
    assert _host_peer_index(info, "h2", 1) == 0
