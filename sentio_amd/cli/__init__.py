"""CLI (reference src/cli capability: sub-commands ingest / api / run;
typer app named `sentio-amd`)."""

from __future__ import annotations

import json
from pathlib import Path

import typer

app = typer.Typer(name="sentio-amd", help="MI355X-native RAG serving engine")
ingest_app = typer.Typer(help="Ingest documents")
api_app = typer.Typer(help="Serve the HTTP API")
bench_app = typer.Typer(help="Benchmarks")
app.add_typer(ingest_app, name="ingest")
app.add_typer(api_app, name="api")
app.add_typer(bench_app, name="bench")


@ingest_app.command("directory")
def ingest_directory_cmd(
    path: str = typer.Argument(..., help="Directory of documents"),
    recursive: bool = typer.Option(True),
):
    from sentio_amd.ingest.ingestor import ingest_directory
    from sentio_amd.serving.container import get_container

    c = get_container()
    result = ingest_directory(path, c.ingestor(), recursive)
    typer.echo(json.dumps(result, indent=2))


@ingest_app.command("file")
def ingest_file_cmd(path: str = typer.Argument(...)):
    from sentio_amd.ingest.readers import read_file
    from sentio_amd.serving.container import get_container

    doc = read_file(Path(path))
    if doc is None:
        typer.echo("unsupported or empty file", err=True)
        raise typer.Exit(1)
    c = get_container()
    result = c.ingestor().ingest_document(doc)
    typer.echo(json.dumps(result, indent=2))


@api_app.command("start")
def api_start(
    host: str = typer.Option(None), port: int = typer.Option(None)
):
    import uvicorn

    from sentio_amd.config import settings
    from sentio_amd.serving.app import create_app

    uvicorn.run(create_app(), host=host or settings.api_host,
                port=port or settings.api_port)


@bench_app.command("run")
def bench_run(steps: int = 5, warmup: int = 1, gpus: int = 1):
    import subprocess
    import sys

    subprocess.run([sys.executable, "bench.py", "--steps", str(steps),
                    "--warmup", str(warmup), "--gpus", str(gpus)], check=True)


index_app = typer.Typer(help="Index snapshots (HBM→disk save / load)")
app.add_typer(index_app, name="index")


@index_app.command("save")
def index_save(directory: str = typer.Argument(...)):
    from sentio_amd.serving.container import get_container

    typer.echo(json.dumps(get_container().save_indexes(directory), indent=2))


@index_app.command("load")
def index_load(directory: str = typer.Argument(...)):
    from sentio_amd.serving.container import get_container

    typer.echo(json.dumps(get_container().load_indexes(directory), indent=2))


@app.command("chat")
def chat_once(question: str = typer.Argument(...)):
    from sentio_amd.serving.container import get_container
    from sentio_amd.serving.handlers import ChatHandler

    c = get_container()
    c.initialize_all()
    result = ChatHandler(c).process(question)
    typer.echo(json.dumps(result, indent=2))


@app.command("run")
def run_all(host: str = typer.Option(None), port: int = typer.Option(None)):
    """Start the full stack: API server (the built-in /ui page is the UI —
    reference `sentio run all` spawned API + Streamlit + Qdrant docker;
    here everything is in-process)."""
    import uvicorn

    from sentio_amd.config import settings
    from sentio_amd.serving.app import create_app

    resolved_port = port or settings.api_port
    typer.echo(f"serving API + UI on http://{host or settings.api_host}:"
               f"{resolved_port} (UI at /ui)")
    uvicorn.run(create_app(), host=host or settings.api_host,
                port=resolved_port)


def main() -> None:
    app()


if __name__ == "__main__":
    main()
