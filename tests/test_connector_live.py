"""Connector lifecycle tests against REAL spawned servers (the reference
only mocked these paths)."""

import socket
import time

import pytest

from fei_amd.tools.memdir_connector import MemdirConnector


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.fixture
def connector(memdir_base):
    port = _free_port()
    conn = MemdirConnector(server_url=f"http://127.0.0.1:{port}",
                           api_key="testkey", base=memdir_base)
    yield conn
    conn.stop_server_command()


def test_spawn_use_stop(connector):
    assert connector.get_server_status()["running"] is False
    out = connector.start_server_command()
    assert out.get("success"), out
    assert connector.check_connection()

    out = connector.create_memory({"Subject": "live test", "Tags": "live"},
                                  "over http", flags="F")
    assert out.get("success"), out
    out = connector.search("#live", with_content=True)
    assert out["count"] == 1
    assert out["results"][0]["content"] == "over http"

    mem_id = out["results"][0]["metadata"]["unique"]
    assert connector.move_memory(mem_id, ".Projects")["success"]
    assert connector.folder_stats(".Projects")["total"] == 1
    assert connector.run_filters()["processed"] >= 0

    out = connector.stop_server_command()
    assert out["success"]
    time.sleep(0.3)
    assert connector.get_server_status()["running"] is False


def test_wrong_api_key_rejected(connector):
    connector.start_server_command()
    bad = MemdirConnector(server_url=connector.server_url, api_key="nope",
                          base=connector.base)
    out = bad.list_memories()
    assert "error" in out


def test_folder_symlink(memdir_base):
    from fei_amd.memdir import utils as mu
    from fei_amd.memdir.folders import MemdirFolderManager
    mgr = MemdirFolderManager(memdir_base)
    mgr.create_folder(".Projects/alpha")
    mu.create_memory(".Projects/alpha", {"Subject": "via link"}, "",
                     base=memdir_base, status="cur")
    assert mgr.link_folder(".Projects/alpha", ".Current")
    assert len(mu.list_memories(".Current", "cur", base=memdir_base)) == 1
    assert mgr.unlink_folder(".Current")
    assert len(mu.list_memories(".Projects/alpha", "cur", base=memdir_base)) == 1
