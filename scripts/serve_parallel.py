"""Thin shim: the distributed-serving entry point lives in the package
(bee2bee_amd/parallel/serve_main.py) so `python -m bee2bee_amd
serve-parallel` and pip installs carry it; this path keeps the documented
`scripts/serve_parallel.py` launch working."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from bee2bee_amd.parallel.serve_main import main

if __name__ == "__main__":
    main()
