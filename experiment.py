#!/usr/bin/env python
"""CLI-compatibility entry point: `python experiment.py COMMAND` dispatches
into the MI355X-native framework (flake16_framework_amd.cli), mirroring the
reference pipeline's command surface."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from flake16_framework_amd.cli import main

if __name__ == "__main__":
    main()
