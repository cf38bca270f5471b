"""`python -m sentio_amd.cli <cmd>` — module entry point for the Typer app
(the reference installed a `sentio` console script, pyproject.toml:116; in
this image nothing is pip-installed, so the runnable form is `-m`)."""

from sentio_amd.cli import main

main()
