"""Store layer tests: LocalStore on a real FS and S3Store against an
in-memory fake boto3 (no network; VERDICT r01 weak #7 — S3Store shipped
untested). The fake implements exactly the client surface S3Store calls:
upload_file/download_file/put_object/get_object/head_object/delete_object
and the list_objects_v2 paginator."""

import io
import sys
import types
from pathlib import Path

import pytest

from photon_amd.fed.store import LocalStore, S3Store, get_store


def test_local_store_roundtrip(tmp_path):
    st = LocalStore(tmp_path)
    st.write_bytes("run/a/x.bin", b"hello")
    assert st.exists("run/a/x.bin")
    assert st.read_bytes("run/a/x.bin") == b"hello"
    src = tmp_path / "src.txt"
    src.write_text("data")
    st.upload(src, "run/b/src.txt")
    st.download("run/b/src.txt", tmp_path / "out.txt")
    assert (tmp_path / "out.txt").read_text() == "data"
    assert st.list("run") == ["run/a/x.bin", "run/b/src.txt"]
    st.delete("run/a")
    assert st.list("run") == ["run/b/src.txt"]


class _FakeS3Client:
    def __init__(self):
        self.objects: dict[tuple, bytes] = {}

    def upload_file(self, src, bucket, key):
        self.objects[(bucket, key)] = Path(src).read_bytes()

    def download_file(self, bucket, key, dst):
        Path(dst).write_bytes(self.objects[(bucket, key)])

    def put_object(self, Bucket, Key, Body):
        self.objects[(Bucket, Key)] = Body

    def get_object(self, Bucket, Key):
        return {"Body": io.BytesIO(self.objects[(Bucket, Key)])}

    def head_object(self, Bucket, Key):
        if (Bucket, Key) not in self.objects:
            raise KeyError(Key)

    def delete_object(self, Bucket, Key):
        self.objects.pop((Bucket, Key), None)

    def get_paginator(self, name):
        assert name == "list_objects_v2"
        objects = self.objects

        class _P:
            def paginate(self, Bucket, Prefix):
                keys = sorted(k for (b, k) in objects if b == Bucket
                              and k.startswith(Prefix))
                yield {"Contents": [{"Key": k} for k in keys]}

        return _P()


@pytest.fixture
def fake_boto3(monkeypatch):
    client = _FakeS3Client()
    mod = types.ModuleType("boto3")
    mod.client = lambda service, endpoint_url=None: client
    monkeypatch.setitem(sys.modules, "boto3", mod)
    return client


def test_s3_store_roundtrip(fake_boto3, tmp_path):
    st = S3Store("bkt", prefix="runs/x")
    st.write_bytes("server/1/state.bin", b"\x01\x02")
    assert st.exists("server/1/state.bin")
    assert not st.exists("server/2/state.bin")
    assert st.read_bytes("server/1/state.bin") == b"\x01\x02"
    src = tmp_path / "p.npz"
    src.write_bytes(b"npz")
    st.upload(src, "server/1/current_server_parameters.npz")
    st.download("server/1/current_server_parameters.npz", tmp_path / "d.npz")
    assert (tmp_path / "d.npz").read_bytes() == b"npz"
    # prefix handling: keys are namespaced under runs/x but listed relative
    assert st.list("server") == [
        "server/1/current_server_parameters.npz",
        "server/1/state.bin",
    ]
    st.delete("server/1")
    assert st.list("server") == []
    assert ("bkt", "runs/x/server/1/state.bin") not in fake_boto3.objects


def test_get_store_dispatch(fake_boto3, tmp_path):
    local = get_store({"photon": {"saving_path": str(tmp_path)}})
    assert isinstance(local, LocalStore)
    s3 = get_store({
        "comm_stack": {"s3": True},
        "s3_comm": {"bucket_name": "bkt", "prefix": "p"},
    })
    assert isinstance(s3, S3Store) and s3.bucket == "bkt"
