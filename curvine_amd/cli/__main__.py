import sys

from curvine_amd.cli.cv import main

sys.exit(main())
