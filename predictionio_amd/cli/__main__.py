from predictionio_amd.cli.main import main

main()
