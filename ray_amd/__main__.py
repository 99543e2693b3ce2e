from ray_amd.scripts import main
import sys
sys.exit(main())
