"""Node error paths: every handler failure maps to a typed ResponseError
(the reference's failure-reporting contract; error cases per
routes.py / uploads.py semantics)."""
import threading

import numpy as np
import pytest

from distributedllm_amd.cluster.client import Connection, OperationFailedError
from distributedllm_amd.cluster.node import NodeServer
from distributedllm_amd.cluster import protocol as P


@pytest.fixture()
def conn(tmp_path):
    srv = NodeServer("127.0.0.1", 0, str(tmp_path / "uploads"))
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    c = Connection("127.0.0.1", srv.port)
    yield c
    c.close()
    srv.shutdown()
    srv.server_close()


def test_load_missing_slice(conn):
    with pytest.raises(OperationFailedError, match="slice_not_found"):
        conn.load_slice("nope.bin")


def test_propagate_without_slice(conn):
    with pytest.raises(OperationFailedError, match="slice_not_loaded"):
        conn.propagate_forward(np.ones((1, 4), np.float32), start_pos=0)


def test_parallel_upload_forbidden(conn):
    first = conn._rpc(P.RequestUploadBegin(kind="slice", metadata="{}"))
    with pytest.raises(OperationFailedError, match="upload_failed"):
        conn._rpc(P.RequestUploadBegin(kind="slice", metadata="{}"))
    # the failed begin aborts the active upload (server-side cleanup);
    # a new upload can start afterwards
    again = conn._rpc(P.RequestUploadBegin(kind="slice", metadata="{}"))
    assert again.upload_id != first.upload_id


def test_upload_part_unknown_id(conn):
    with pytest.raises(OperationFailedError, match="upload_failed"):
        conn._rpc(P.RequestUploadPart(upload_id=999, data=b"zz"))


def test_upload_checksum_mismatch(conn):
    begin = conn._rpc(P.RequestUploadBegin(kind="slice",
                                           metadata='{"name": "x.bin"}'))
    conn._rpc(P.RequestUploadPart(upload_id=begin.upload_id, data=b"abcd"))
    with pytest.raises(OperationFailedError, match="upload_failed"):
        conn._rpc(P.RequestUploadEnd(upload_id=begin.upload_id,
                                     total_size=4, checksum="0" * 64))
    # failed upload is not listed
    assert all(e["name"] != "x.bin" for e in conn.list_slices())


def test_dummy_slice_bad_file(conn, tmp_path):
    path = tmp_path / "bad.bin"
    path.write_bytes(b"\x00" * 3)  # not two f32s
    conn.push_file(str(path), "slice", {"name": "bad.bin",
                                        "format": "test"})
    with pytest.raises(OperationFailedError, match="slice_load_error"):
        conn.load_slice("bad.bin")


def test_unknown_message(conn):
    # a node (not a proxy) has no greeting handler -> typed error
    with pytest.raises(OperationFailedError, match="unknown_request"):
        conn._rpc(P.RequestGreeting(name="x"))
