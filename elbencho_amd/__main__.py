"""python -m elbencho_amd — CLI entry point."""

import sys

from elbencho_amd.cli import main

if __name__ == "__main__":
    sys.exit(main())
