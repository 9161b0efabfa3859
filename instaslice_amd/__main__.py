from instaslice_amd.cli import main

raise SystemExit(main())
