from deepconsensus_amd.cli import main

main()
