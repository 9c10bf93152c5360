#!/usr/bin/env python3
"""CLI entry point: python main_al.py <flags> (same surface as the reference's
src/main_al.py; see active_learning_amd/cli.py for the flag list)."""

from active_learning_amd.cli import get_args
from active_learning_amd.main_al import main

if __name__ == "__main__":
    main(get_args())
