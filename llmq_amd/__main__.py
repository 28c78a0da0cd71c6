from llmq_amd.cli.main import main

if __name__ == "__main__":
    main()
