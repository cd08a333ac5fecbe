from fei_amd.ui.cli import main as cli_main

__all__ = ["cli_main"]
