"""``python -m fei_amd`` — CLI entry (reference: fei/__main__.py:10-27,
including the --textual switch, which fei_amd.ui.cli handles)."""

from fei_amd.ui.cli import main

if __name__ == "__main__":
    raise SystemExit(main())
