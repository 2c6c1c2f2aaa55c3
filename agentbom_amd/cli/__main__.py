"""``python -m agentbom_amd.cli`` entry point."""

from agentbom_amd.cli import main

if __name__ == "__main__":
    main()
