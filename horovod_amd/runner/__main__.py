import sys

from horovod_amd.runner.launch import main

sys.exit(main())
