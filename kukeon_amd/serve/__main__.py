from kukeon_amd.serve.server import main

main()
