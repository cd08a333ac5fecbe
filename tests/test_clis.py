"""CLI surfaces: memdir CLI/__main__, memorychain CLI (offline paths),
fei subcommands beyond those covered in test_ui."""

import json

import pytest

from fei_amd.memdir.cli import main as memdir_main
from fei_amd.memdir.__main__ import main as memdir_dunder_main
from fei_amd.memorychain.cli import main as chain_main


def test_memdir_cli_roundtrip(memdir_base, capsys):
    rc = memdir_main(["--base", memdir_base, "create", "--subject",
                      "cli memo", "--tags", "cli", "--flags", "F",
                      "--body", "made by the cli"])
    assert rc == 0
    filename = capsys.readouterr().out.strip()
    assert ":2,F" in filename

    rc = memdir_main(["--base", memdir_base, "list", "--status", "new"])
    assert rc == 0
    assert "cli memo" in capsys.readouterr().out

    unique = filename.split(".")[1]
    rc = memdir_main(["--base", memdir_base, "view", unique])
    assert rc == 0
    assert "made by the cli" in capsys.readouterr().out

    rc = memdir_main(["--base", memdir_base, "move", unique, ".Projects"])
    assert rc == 0
    rc = memdir_main(["--base", memdir_base, "search", "#cli", "--format",
                      "compact"])
    assert rc == 0
    assert "cli memo" in capsys.readouterr().out

    rc = memdir_main(["--base", memdir_base, "flag", unique, "SP"])
    assert rc == 0
    rc = memdir_main(["--base", memdir_base, "folders"])
    assert rc == 0
    assert ".Projects" in capsys.readouterr().out


def test_memdir_dunder_main(memdir_base, monkeypatch, capsys):
    monkeypatch.setenv("MEMDIR_BASE", memdir_base)
    assert memdir_dunder_main(["init-samples", "5"]) == 0
    assert "created 5" in capsys.readouterr().out
    assert memdir_dunder_main(["run-filters"]) == 0
    capsys.readouterr()
    assert memdir_dunder_main(["maintenance"]) == 0
    out = capsys.readouterr().out
    assert "archived" in out


def test_memorychain_cli_start_offline_commands(tmp_path, monkeypatch, capsys):
    """Offline commands against an unreachable node return errors, not
    tracebacks."""
    monkeypatch.setenv("MEMORYCHAIN_NODE", "127.0.0.1:1")   # nothing there
    rc = chain_main(["status"])
    out = capsys.readouterr().out
    assert rc == 1 and "error" in out

    rc = chain_main(["validate"])
    out = capsys.readouterr().out
    assert rc == 1 or "false" in out.lower()


def test_memorychain_cli_against_test_node(tmp_path, monkeypatch, capsys):
    """Drive the CLI's connector against a live in-thread node."""
    import socket
    import threading
    import time
    from fei_amd.memorychain.chain import MemoryChain
    from fei_amd.memorychain.node import MemorychainNode
    from fei_amd.memorychain.wallet import FeiCoinWallet

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    chain = MemoryChain(node_id="clinode", path=str(tmp_path / "c.json"),
                        difficulty=1,
                        wallet=FeiCoinWallet(path=str(tmp_path / "w.json")))
    node = MemorychainNode(node_id="clinode", port=port, chain=chain)
    t = threading.Thread(target=node.run, daemon=True)
    t.start()
    time.sleep(0.8)
    capsys.readouterr()                      # drop the werkzeug banner

    def last_json(text):
        return json.loads(text[text.index("{"):])

    monkeypatch.setenv("MEMORYCHAIN_NODE", f"127.0.0.1:{port}")
    rc = chain_main(["propose", "--subject", "via cli", "--tags", "t"])
    out = last_json(capsys.readouterr().out)
    assert rc == 0 and out["accepted"]

    rc = chain_main(["list"])
    assert "via cli" in capsys.readouterr().out

    rc = chain_main(["task", "--subject", "cli task", "--reward", "2"])
    assert last_json(capsys.readouterr().out)["accepted"]
    rc = chain_main(["tasks"])
    assert "cli task" in capsys.readouterr().out

    rc = chain_main(["validate"])
    assert last_json(capsys.readouterr().out)["valid"]

    rc = chain_main(["status"])
    assert last_json(capsys.readouterr().out)["node_id"] == "clinode"


def test_fei_history_and_mcp_subcommands(tmp_path, monkeypatch, capsys):
    monkeypatch.setenv("HOME", str(tmp_path))
    import fei_amd.ui.cli as cli_mod
    monkeypatch.setattr(cli_mod, "HISTORY_PATH",
                        str(tmp_path / ".fei" / "history.json"))
    h = cli_mod.ChatHistory(str(tmp_path / ".fei" / "history.json"))
    h.add("remembered prompt", "resp")
    assert cli_mod.main(["history"]) == 0
    assert "remembered prompt" in capsys.readouterr().out

    monkeypatch.setenv("FEI_MCP_SERVER_DEMO", "http://localhost:9/rpc")
    assert cli_mod.main(["mcp"]) == 0
    assert "demo" in capsys.readouterr().out


def test_package_import_is_torch_free():
    """`import fei_amd` must stay light (agent/memdir layers don't need
    torch; it costs ~1.5 s) — engine exports are lazy."""
    import subprocess
    import sys
    r = subprocess.run(
        [sys.executable, "-c",
         "import sys, fei_amd; assert 'torch' not in sys.modules; "
         "fei_amd.LocalEngine; assert 'torch' in sys.modules"],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr


def test_memdir_cli_fts(memdir_base, capsys):
    from fei_amd.memdir import utils as mu
    from fei_amd.memdir.cli import main
    mu.create_memory("", {"Subject": "fts cli target"},
                     "zebra content here", status="cur", base=memdir_base)
    assert main(["--base", memdir_base, "index"]) == 0
    assert "indexed 1" in capsys.readouterr().out
    assert main(["--base", memdir_base, "search", "--fts", "zebra"]) == 0
    assert "fts cli target" in capsys.readouterr().out


@pytest.mark.parametrize("mod", [
    "fei_amd", "fei_amd.memdir.cli", "fei_amd.memorychain.cli",
    "fei_amd.serve.api", "fei_amd.memdir.run_server",
])
def test_cli_help_surfaces(mod):
    """Every console entry point parses --help without importing torch-heavy
    or crashing (first thing a new user runs)."""
    import subprocess
    import sys
    r = subprocess.run([sys.executable, "-m", mod, "--help"],
                       capture_output=True, text=True, timeout=90)
    assert r.returncode == 0, r.stderr[-500:]
