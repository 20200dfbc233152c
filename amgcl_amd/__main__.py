"""`python -m amgcl_amd` — the solver CLI (parity: examples/solver.cpp)."""
import sys

from .cli import main

sys.exit(main())
