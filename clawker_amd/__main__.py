"""`python -m clawker_amd` == the clawker CLI."""
import sys

from clawker_amd.cli.root import main

sys.exit(main())
