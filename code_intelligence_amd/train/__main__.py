from .train_cli import main

main()
