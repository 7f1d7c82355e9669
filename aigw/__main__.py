from aigw.cli.main import main

raise SystemExit(main())
