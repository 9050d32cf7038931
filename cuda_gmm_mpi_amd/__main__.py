"""``python -m cuda_gmm_mpi_amd`` — same CLI as the ``gmm`` console script."""
import sys

from .cli import main

if __name__ == "__main__":
    sys.exit(main())
