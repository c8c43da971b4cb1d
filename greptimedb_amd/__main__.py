from greptimedb_amd.cli import main

main()
