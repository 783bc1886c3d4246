from skypilot_amd.cli.main import main

main()
