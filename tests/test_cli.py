"""CLI coverage (reference src/tests had no CLI tests; typer's CliRunner
makes them cheap here)."""

from __future__ import annotations

import json

from typer.testing import CliRunner

from sentio_amd.cli import app

runner = CliRunner()


def _mock_env(monkeypatch, tmp_path):
    monkeypatch.setenv("MOCK_COMPUTE", "1")
    monkeypatch.setenv("SENTIO_DEVICE", "cpu")
    from sentio_amd.config import reload_settings
    from sentio_amd.serving.container import reset_container

    reload_settings()
    reset_container()


def test_cli_ingest_directory_and_chat(monkeypatch, tmp_path):
    _mock_env(monkeypatch, tmp_path)
    (tmp_path / "doc.txt").write_text("the MI355X GPU has 288 GB HBM3E " * 10)
    r = runner.invoke(app, ["ingest", "directory", str(tmp_path)])
    assert r.exit_code == 0, r.output
    stats = json.loads(r.output)
    assert stats["documents"] == 1 and stats["chunks"] >= 1

    r2 = runner.invoke(app, ["chat", "how much memory does MI355X have?"])
    assert r2.exit_code == 0, r2.output
    out = json.loads(r2.output)
    assert out["answer"]


def test_cli_index_save_load(monkeypatch, tmp_path):
    _mock_env(monkeypatch, tmp_path)
    (tmp_path / "d.md").write_text("retrieval augmented generation " * 20)
    runner.invoke(app, ["ingest", "directory", str(tmp_path)])
    snap = tmp_path / "snap"
    r = runner.invoke(app, ["index", "save", str(snap)])
    assert r.exit_code == 0, r.output
    assert (snap / "dense.pt").exists() and (snap / "bm25.npz").exists()

    from sentio_amd.serving.container import get_container, reset_container
    reset_container()
    r2 = runner.invoke(app, ["index", "load", str(snap)])
    assert r2.exit_code == 0, r2.output
    assert json.loads(r2.output)["docs"] >= 1


def test_cli_help_lists_subcommands():
    r = runner.invoke(app, ["--help"])
    assert r.exit_code == 0
    for sub in ("ingest", "api", "bench", "index", "chat", "run"):
        assert sub in r.output


def test_cli_api_start_and_run_invoke_uvicorn(monkeypatch):
    """`api start` and `run` build the app and hand it to uvicorn with the
    resolved host/port (uvicorn itself is stubbed)."""
    import uvicorn
    from typer.testing import CliRunner

    from sentio_amd.cli import app as cli_app

    calls = []
    monkeypatch.setenv("MOCK_COMPUTE", "true")
    monkeypatch.setenv("SENTIO_DEVICE", "cpu")
    monkeypatch.setattr(uvicorn, "run",
                        lambda a, host, port: calls.append((host, port)))
    runner = CliRunner()
    r1 = runner.invoke(cli_app, ["api", "start", "--port", "9911"])
    assert r1.exit_code == 0, r1.output
    r2 = runner.invoke(cli_app, ["run", "--host", "127.0.0.1", "--port", "9912"])
    assert r2.exit_code == 0, r2.output
    assert ("UI at /ui") in r2.output
    assert calls == [("0.0.0.0", 9911), ("127.0.0.1", 9912)]
