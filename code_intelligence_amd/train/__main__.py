from .train_cli import main

if __name__ == "__main__":  # python -m code_intelligence_amd.train
    main()
