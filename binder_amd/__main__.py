"""python -m binder_amd -> operator CLI."""
import sys

from .cli import main

sys.exit(main())
