"""``python -m fei_amd.memdir`` dispatcher
(reference parity: memdir_tools/__main__.py:10-90): routes maintenance and
sample-generation commands, defers everything else to the CLI."""

from __future__ import annotations

import sys


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    if argv and argv[0] == "init-samples":
        from fei_amd.memdir.create_samples import create_samples
        n = create_samples(count=int(argv[1]) if len(argv) > 1 else 25)
        print(f"created {n} sample memories")
        return 0
    if argv and argv[0] == "run-filters":
        from fei_amd.memdir.filter import run_filters
        report = run_filters()
        print(f"processed {report['processed']}, actions: {len(report['actions'])}")
        return 0
    if argv and argv[0] == "maintenance":
        from fei_amd.memdir.archiver import MemoryArchiver
        print(MemoryArchiver().run_maintenance())
        return 0
    from fei_amd.memdir.cli import main as cli_main
    return cli_main(argv)


if __name__ == "__main__":
    raise SystemExit(main())
