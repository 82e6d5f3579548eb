from dts_amd.server.app import main

main()
