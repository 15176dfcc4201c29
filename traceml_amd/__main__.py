"""``python -m traceml_amd`` == the traceml-amd CLI."""

import sys

from traceml_amd.launcher.cli import main

sys.exit(main())
