from kukeon_amd.cli.main import main

main()
